"""REST data-plane tests with a DummyModel (style mirrors reference
python/kserve/test/test_server.py:146 — fresh code)."""

import json

import numpy as np
import pytest
from fastapi.testclient import TestClient

from kserve_amd import constants
from kserve_amd.model import Model
from kserve_amd.model_repository import ModelRepository
from kserve_amd.protocol.dataplane import DataPlane
from kserve_amd.protocol.infer_type import (
    InferInput,
    InferOutput,
    InferRequest,
    InferResponse,
)
from kserve_amd.protocol.rest.server import create_app


class DummyModel(Model):
    def __init__(self, name="dummy"):
        super().__init__(name)
        self.ready = True

    def predict(self, payload, headers=None):
        if isinstance(payload, InferRequest):
            x = payload.inputs[0].as_numpy()
            out = InferOutput(
                "output-0", list(x.shape), payload.inputs[0].datatype
            )
            out.set_data_from_numpy(x * 2, binary_data=payload.inputs[0].raw_data is not None)
            return InferResponse(payload.id, self.name, [out])
        instances = payload["instances"]
        return {"predictions": [[v * 2 for v in row] for row in instances]}

    def explain(self, payload, headers=None):
        return {"explanations": "dummy"}


@pytest.fixture
def client():
    repo = ModelRepository()
    repo.update(DummyModel())
    app = create_app(DataPlane(repo))
    return TestClient(app)


class TestV1:
    def test_list_models(self, client):
        r = client.get("/v1/models")
        assert r.status_code == 200
        assert r.json() == {"models": ["dummy"]}

    def test_model_ready(self, client):
        r = client.get("/v1/models/dummy")
        assert r.status_code == 200
        assert r.json()["ready"] is True

    def test_model_not_found(self, client):
        r = client.get("/v1/models/nope")
        assert r.status_code == 404

    def test_predict(self, client):
        r = client.post(
            "/v1/models/dummy:predict", json={"instances": [[1, 2], [3, 4]]}
        )
        assert r.status_code == 200
        assert r.json() == {"predictions": [[2, 4], [6, 8]]}

    def test_explain(self, client):
        r = client.post(
            "/v1/models/dummy:explain", json={"instances": [[1, 2]]}
        )
        assert r.status_code == 200
        assert r.json()["explanations"] == "dummy"

    def test_invalid_instances(self, client):
        r = client.post(
            "/v1/models/dummy:predict", json={"instances": "notalist"}
        )
        assert r.status_code == 400


class TestV2:
    def test_metadata(self, client):
        r = client.get("/v2")
        assert r.status_code == 200
        assert "binary_tensor_data_extension" in r.json()["extensions"]

    def test_health(self, client):
        assert client.get("/v2/health/live").json()["live"] is True
        assert client.get("/v2/health/ready").json()["ready"] is True

    def test_model_metadata(self, client):
        r = client.get("/v2/models/dummy")
        assert r.status_code == 200
        assert r.json()["name"] == "dummy"

    def test_infer_json(self, client):
        req = {
            "inputs": [
                {"name": "x", "shape": [2, 2], "datatype": "FP32", "data": [1, 2, 3, 4]}
            ]
        }
        r = client.post("/v2/models/dummy/infer", json=req)
        assert r.status_code == 200
        body = r.json()
        assert body["model_name"] == "dummy"
        assert body["outputs"][0]["data"] == [2.0, 4.0, 6.0, 8.0]

    def test_infer_binary(self, client):
        x = np.arange(4, dtype=np.float32).reshape(2, 2)
        inp = InferInput("x", [2, 2], "FP32")
        inp.set_data_from_numpy(x, binary_data=True)
        req = InferRequest("dummy", [inp])
        body, json_len = req.to_rest()
        r = client.post(
            "/v2/models/dummy/infer",
            content=body,
            headers={
                constants.INFERENCE_CONTENT_LENGTH_HEADER: str(json_len),
                "content-type": "application/octet-stream",
            },
        )
        assert r.status_code == 200
        resp_len = int(r.headers[constants.INFERENCE_CONTENT_LENGTH_HEADER])
        decoded = InferResponse.from_bytes(r.content, resp_len)
        np.testing.assert_array_equal(decoded.outputs[0].as_numpy(), x * 2)

    def test_infer_model_not_found(self, client):
        r = client.post(
            "/v2/models/nope/infer",
            json={"inputs": [{"name": "x", "shape": [1], "datatype": "FP32", "data": [1]}]},
        )
        assert r.status_code == 404

    def test_repository_index(self, client):
        r = client.post("/v2/repository/index")
        assert r.status_code == 200
        assert r.json()[0]["name"] == "dummy"
        assert r.json()[0]["state"] == "READY"


class TestMetrics:
    def test_prometheus_endpoint(self, client):
        client.post("/v1/models/dummy:predict", json={"instances": [[1]]})
        r = client.get("/metrics")
        assert r.status_code == 200
        assert b"request_predict_seconds" in r.content
