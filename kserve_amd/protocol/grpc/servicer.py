"""gRPC V2 inference servicer.

Reference parity: python/kserve protocol/grpc/servicer.py:26-109 — delegates to
DataPlane; tensor conversion between ModelInferRequest/Response protos and
InferRequest/InferResponse.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import grpc
import numpy as np

from kserve_amd.errors import InvalidInput, ModelNotFound, ModelNotReady
from kserve_amd.logging import logger
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.grpc import proto
from kserve_amd.protocol.infer_type import (
    InferInput,
    InferOutput,
    InferRequest,
    InferResponse,
    RequestedOutput,
    deserialize_bytes_tensor,
    serialize_bytes_tensor,
    to_np_dtype,
)

# datatype -> InferTensorContents field
_CONTENT_FIELD = {
    "BOOL": "bool_contents",
    "INT8": "int_contents",
    "INT16": "int_contents",
    "INT32": "int_contents",
    "INT64": "int64_contents",
    "UINT8": "uint_contents",
    "UINT16": "uint_contents",
    "UINT32": "uint_contents",
    "UINT64": "uint64_contents",
    "FP32": "fp32_contents",
    "FP64": "fp64_contents",
    "BYTES": "bytes_contents",
}


def _params_to_dict(pmap) -> Dict:
    out = {}
    for k, v in pmap.items():
        which = v.WhichOneof("parameter_choice")
        out[k] = getattr(v, which) if which else None
    return out


def _dict_to_params(d: Dict, pmap):
    for k, v in (d or {}).items():
        if isinstance(v, bool):
            pmap[k].bool_param = v
        elif isinstance(v, int):
            pmap[k].int64_param = v
        elif isinstance(v, float):
            pmap[k].double_param = v
        elif v is not None:
            pmap[k].string_param = str(v)


def grpc_request_to_infer_request(req) -> InferRequest:
    """ModelInferRequest proto -> InferRequest (reference infer_type.py:548)."""
    inputs: List[InferInput] = []
    raw = list(req.raw_input_contents)
    use_raw = len(raw) > 0
    for idx, t in enumerate(req.inputs):
        params = _params_to_dict(t.parameters)
        inp = InferInput(
            name=t.name, shape=list(t.shape), datatype=t.datatype, parameters=params
        )
        if use_raw:
            if idx >= len(raw):
                raise InvalidInput(
                    f"raw_input_contents missing tensor for input {t.name}"
                )
            inp.set_raw_data(raw[idx])
        else:
            field = _CONTENT_FIELD.get(t.datatype)
            if field is None:
                raise InvalidInput(
                    f"{t.datatype} gRPC input {t.name} requires raw_input_contents"
                )
            inp.data = list(getattr(t.contents, field))
        inputs.append(inp)
    request_outputs = [
        RequestedOutput(o.name, _params_to_dict(o.parameters)) for o in req.outputs
    ]
    return InferRequest(
        model_name=req.model_name,
        infer_inputs=inputs,
        request_id=req.id or None,
        request_outputs=request_outputs or None,
        parameters=_params_to_dict(req.parameters),
        from_grpc=True,
    )


def infer_response_to_grpc(resp: InferResponse):
    """InferResponse -> ModelInferResponse proto; tensors go out raw."""
    out = proto.ModelInferResponse()
    out.model_name = resp.model_name
    out.id = resp.id or ""
    if resp.model_version:
        out.model_version = resp.model_version
    _dict_to_params(resp.parameters, out.parameters)
    for o in resp.outputs:
        t = out.outputs.add()
        t.name = o.name
        t.datatype = o.datatype
        t.shape.extend(int(s) for s in o.shape)
        _dict_to_params(
            {k: v for k, v in o.parameters.items() if k != "binary_data_size"},
            t.parameters,
        )
        if o.raw_data is not None:
            out.raw_output_contents.append(o.raw_data)
        else:
            arr = o.as_numpy()
            if o.datatype == "BYTES":
                out.raw_output_contents.append(serialize_bytes_tensor(arr))
            else:
                out.raw_output_contents.append(np.ascontiguousarray(arr).tobytes())
    return out


def dict_response_to_grpc(model_name: str, result: Dict):
    """V1-style dict prediction -> proto via a single BYTES json tensor."""
    import json

    out = proto.ModelInferResponse()
    out.model_name = model_name
    t = out.outputs.add()
    t.name = "output-0"
    t.datatype = "BYTES"
    t.shape.extend([1])
    out.raw_output_contents.append(
        serialize_bytes_tensor(np.array([json.dumps(result).encode()], dtype=np.object_))
    )
    return out


class InferenceServicer:
    def __init__(self, dataplane: DataPlane, model_repository_extension=None):
        self.dataplane = dataplane
        self.model_repository_extension = model_repository_extension

    async def ServerLive(self, request, context):
        return proto.ServerLiveResponse(live=await self.dataplane.live())

    async def ServerReady(self, request, context):
        return proto.ServerReadyResponse(ready=await self.dataplane.ready())

    async def ModelReady(self, request, context):
        try:
            ready = await self.dataplane.model_ready(request.name)
        except ModelNotFound:
            await context.abort(
                grpc.StatusCode.NOT_FOUND, f"Model {request.name} not found"
            )
        return proto.ModelReadyResponse(ready=ready)

    async def ServerMetadata(self, request, context):
        md = await self.dataplane.metadata()
        return proto.ServerMetadataResponse(
            name=md["name"], version=md["version"], extensions=md["extensions"]
        )

    async def ModelMetadata(self, request, context):
        try:
            md = await self.dataplane.model_metadata(request.name)
        except ModelNotFound:
            await context.abort(
                grpc.StatusCode.NOT_FOUND, f"Model {request.name} not found"
            )
        resp = proto.ModelMetadataResponse()
        resp.name = md["name"]
        resp.platform = md.get("platform", "")
        for io_key, field in (("inputs", resp.inputs), ("outputs", resp.outputs)):
            for t in md.get(io_key) or []:
                tm = field.add()
                if isinstance(t, dict):
                    tm.name = t.get("name", "")
                    tm.datatype = t.get("datatype", "")
                    tm.shape.extend(int(s) for s in t.get("shape", []))
        return resp

    async def ModelInfer(self, request, context):
        headers = dict(context.invocation_metadata()) if context else {}
        try:
            infer_request = grpc_request_to_infer_request(request)
            result = await self.dataplane.infer(
                request.model_name, infer_request, headers
            )
        except InvalidInput as e:
            await context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
        except ModelNotFound as e:
            await context.abort(grpc.StatusCode.NOT_FOUND, str(e))
        except ModelNotReady as e:
            await context.abort(grpc.StatusCode.UNAVAILABLE, str(e))
        if isinstance(result, InferResponse):
            return infer_response_to_grpc(result)
        if isinstance(result, dict):
            return dict_response_to_grpc(request.model_name, result)
        await context.abort(
            grpc.StatusCode.INTERNAL, f"Unexpected result type {type(result)}"
        )

    async def RepositoryIndex(self, request, context):
        resp = proto.RepositoryIndexResponse()
        for entry in self.dataplane.model_registry.index():
            mi = resp.models.add()
            mi.name = entry["name"]
            mi.state = entry["state"]
            mi.reason = entry.get("reason", "")
        return resp

    async def RepositoryModelLoad(self, request, context):
        if self.model_repository_extension is not None:
            await self.model_repository_extension.load(request.model_name)
        else:
            ok = self.dataplane.model_registry.load(request.model_name)
            if not ok:
                await context.abort(
                    grpc.StatusCode.NOT_FOUND,
                    f"Model {request.model_name} not found",
                )
        return proto.RepositoryModelLoadResponse()

    async def RepositoryModelUnload(self, request, context):
        try:
            if self.model_repository_extension is not None:
                await self.model_repository_extension.unload(request.model_name)
            else:
                self.dataplane.model_registry.unload(request.model_name)
        except ModelNotFound:
            await context.abort(
                grpc.StatusCode.NOT_FOUND, f"Model {request.model_name} not found"
            )
        return proto.RepositoryModelUnloadResponse()
