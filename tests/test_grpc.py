"""gRPC V2 servicer tests over a real in-process grpc.aio server."""

import asyncio

import grpc
import numpy as np
import pytest

from kserve_amd.model import Model
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.grpc import proto
from kserve_amd.protocol.grpc.server import GRPCServer
from kserve_amd.protocol.infer_type import InferOutput, InferRequest, InferResponse


class EchoTwice(Model):
    def __init__(self):
        super().__init__("echo")
        self.ready = True

    def predict(self, payload: InferRequest, headers=None):
        x = payload.inputs[0].as_numpy()
        out = InferOutput("output-0", list(x.shape), payload.inputs[0].datatype)
        out.set_data_from_numpy(x * 2, binary_data=True)
        return InferResponse(payload.id, self.name, [out])


@pytest.fixture
def dataplane():
    repo = ModelRepository()
    repo.update(EchoTwice())
    return DataPlane(repo)


def _run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_grpc_end_to_end(dataplane):
    async def main():
        server = GRPCServer(dataplane, port=0)
        start_task = asyncio.create_task(server.start())
        await asyncio.sleep(0.2)
        port = server.bound_port
        async with grpc.aio.insecure_channel(f"127.0.0.1:{port}") as ch:
            # health
            live = ch.unary_unary(
                f"/{proto.SERVICE_NAME}/ServerLive",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=proto.ServerLiveResponse.FromString,
            )
            resp = await live(proto.ServerLiveRequest())
            assert resp.live is True

            ready = ch.unary_unary(
                f"/{proto.SERVICE_NAME}/ModelReady",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=proto.ModelReadyResponse.FromString,
            )
            resp = await ready(proto.ModelReadyRequest(name="echo"))
            assert resp.ready is True

            # infer with raw contents
            infer = ch.unary_unary(
                f"/{proto.SERVICE_NAME}/ModelInfer",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=proto.ModelInferResponse.FromString,
            )
            req = proto.ModelInferRequest()
            req.model_name = "echo"
            t = req.inputs.add()
            t.name = "x"
            t.datatype = "FP32"
            t.shape.extend([2, 2])
            x = np.arange(4, dtype=np.float32)
            req.raw_input_contents.append(x.tobytes())
            resp = await infer(req)
            assert resp.model_name == "echo"
            y = np.frombuffer(resp.raw_output_contents[0], dtype=np.float32)
            np.testing.assert_array_equal(y, x * 2)

            # infer with typed contents
            req2 = proto.ModelInferRequest()
            req2.model_name = "echo"
            t = req2.inputs.add()
            t.name = "x"
            t.datatype = "FP32"
            t.shape.extend([2])
            t.contents.fp32_contents.extend([1.0, 2.0])
            resp2 = await infer(req2)
            y2 = np.frombuffer(resp2.raw_output_contents[0], dtype=np.float32)
            np.testing.assert_array_equal(y2, np.array([2.0, 4.0], dtype=np.float32))

            # repository index
            index = ch.unary_unary(
                f"/{proto.SERVICE_NAME}/RepositoryIndex",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=proto.RepositoryIndexResponse.FromString,
            )
            resp3 = await index(proto.RepositoryIndexRequest())
            assert resp3.models[0].name == "echo"

            # unknown model -> NOT_FOUND
            req4 = proto.ModelInferRequest()
            req4.model_name = "missing"
            with pytest.raises(grpc.aio.AioRpcError) as e:
                await infer(req4)
            assert e.value.code() == grpc.StatusCode.NOT_FOUND

        await server.stop(grace=0.1)
        start_task.cancel()

    _run(main())


def test_grpc_client_roundtrip(dataplane):
    """InferenceGRPCClient against the in-process server."""
    async def main():
        from kserve_amd.inference_client import InferenceGRPCClient
        from kserve_amd.protocol.grpc.server import GRPCServer
        from kserve_amd.protocol.infer_type import InferInput, InferRequest

        server = GRPCServer(dataplane, port=0)
        task = asyncio.create_task(server.start())
        await asyncio.sleep(0.2)
        client = InferenceGRPCClient(f"127.0.0.1:{server.bound_port}")
        assert await client.is_server_ready()
        assert await client.is_model_ready("echo")
        x = np.arange(6, dtype=np.float32).reshape(2, 3)
        inp = InferInput("x", [2, 3], "FP32")
        inp.set_data_from_numpy(x, binary_data=True)
        resp = await client.infer(InferRequest("echo", [inp]))
        np.testing.assert_array_equal(resp.outputs[0].as_numpy(), x * 2)
        await client.close()
        await server.stop(grace=0.1)
        task.cancel()

    _run(main())
