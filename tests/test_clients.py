"""Inference clients (REST V1/V2 incl. binary) + qpext aggregator."""

import asyncio
import json

import httpx
import numpy as np
import pytest
from fastapi.testclient import TestClient

from kserve_amd import constants
from kserve_amd.inference_client import InferenceRESTClient, RESTConfig
from kserve_amd.model import Model
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.infer_type import InferInput, InferRequest, InferResponse
from kserve_amd.protocol.rest.server import create_app


class Doubler(Model):
    def __init__(self):
        super().__init__("d")
        self.ready = True

    def predict(self, payload, headers=None):
        if isinstance(payload, InferRequest):
            x = payload.inputs[0].as_numpy()
            from kserve_amd.protocol.infer_type import InferOutput

            o = InferOutput("output-0", list(x.shape), payload.inputs[0].datatype)
            o.set_data_from_numpy(x * 2, binary_data=payload.inputs[0].raw_data is not None)
            return InferResponse(payload.id, self.name, [o])
        return {"predictions": [v * 2 for v in payload["instances"]]}


def make_transport():
    repo = ModelRepository()
    repo.update(Doubler())
    app = create_app(DataPlane(repo))
    return httpx.ASGITransport(app=app)


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


class TestRESTClient:
    def test_v1_predict(self):
        client = InferenceRESTClient(
            RESTConfig(protocol="v1", transport=make_transport())
        )
        out = run(client.infer("http://srv", {"instances": [1, 2]}, "d"))
        assert out["predictions"] == [2, 4]

    def test_v2_binary_roundtrip(self):
        client = InferenceRESTClient(
            RESTConfig(protocol="v2", transport=make_transport())
        )
        x = np.arange(6, dtype=np.float32).reshape(2, 3)
        inp = InferInput("x", [2, 3], "FP32")
        inp.set_data_from_numpy(x, binary_data=True)
        resp = run(client.infer("http://srv", InferRequest("d", [inp]), "d"))
        assert isinstance(resp, InferResponse)
        np.testing.assert_array_equal(resp.outputs[0].as_numpy(), x * 2)

    def test_health(self):
        client = InferenceRESTClient(RESTConfig(transport=make_transport()))
        assert run(client.is_server_live("http://srv"))
        assert run(client.is_server_ready("http://srv"))
        assert run(client.is_model_ready("http://srv", "d"))

    def test_error_raises(self):
        from kserve_amd.errors import InferenceError

        client = InferenceRESTClient(
            RESTConfig(protocol="v1", transport=make_transport())
        )
        with pytest.raises(InferenceError):
            run(client.infer("http://srv", {"instances": [1]}, "missing"))


class TestQpext:
    def test_merge(self):
        from kserve_amd.agent.qpext import create_qpext_app

        def target(request: httpx.Request) -> httpx.Response:
            if request.url.port == 9091:
                return httpx.Response(
                    200, text="queue_requests_total 5\npython_gc_count 3\n"
                )
            return httpx.Response(200, text="request_predict_seconds_count 7\n")

        app = create_qpext_app(
            ["http://127.0.0.1:9091/metrics", "http://127.0.0.1:8080/metrics"],
            transport=httpx.MockTransport(target),
        )
        with TestClient(app) as c:
            r = c.get("/metrics")
        assert "queue_requests_total 5" in r.text
        assert "request_predict_seconds_count 7" in r.text
        assert "python_gc_count" not in r.text
