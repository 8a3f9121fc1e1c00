"""Platform e2e: CR → live controller → rendered pod → ACTUAL model server
answering the rendered route.

The reference's e2e suite (test/e2e/predictor/test_sklearn.py) applies an
InferenceService to a real cluster and predicts through the ingress. This
is that loop without kubelet: the live controller converges the CR on the
fake API server, the test then BOOTS the model server the rendered
Deployment names (same module, same args contract) in-process and serves
the prediction; the graph test does the same for an InferenceGraph via
the router."""

import json

import numpy as np
import pytest
from fastapi.testclient import TestClient

from kserve_amd.controlplane.apiserver import FakeAPIServer
from kserve_amd.controlplane.crd_controllers import InferenceGraphController
from kserve_amd.controlplane.isvc_controller import (
    FakeDeploymentController,
    InferenceServiceController,
)
from tests.test_live_controller import converge, make_isvc


@pytest.fixture(scope="module")
def iris_dir(tmp_path_factory):
    sklearn = pytest.importorskip("sklearn")
    import joblib
    from sklearn.datasets import load_iris
    from sklearn.linear_model import LogisticRegression

    d = tmp_path_factory.mktemp("iris-e2e")
    X, y = load_iris(return_X_y=True)
    joblib.dump(LogisticRegression(max_iter=200).fit(X, y), d / "model.joblib")
    return str(d)


def boot_rendered_server(dep_manifest, model_dir):
    """Start the model server the rendered Deployment names, in-process:
    resolve `python -m <module>` from the container command, honor the
    --model_name arg, point --model_dir at the downloaded artifacts (the
    storage-initializer's /mnt/models role)."""
    c = dep_manifest["spec"]["template"]["spec"]["containers"][0]
    assert c["command"][:2] == ["python", "-m"]
    module = c["command"][2]
    assert module == "kserve_amd.runtimes.sklearnserver"
    args = {
        a.split("=", 1)[0]: a.split("=", 1)[1]
        for a in c["args"]
        if "=" in a
    }
    name = args["--model_name"]
    from kserve_amd.model_repository import ModelRepository
    from kserve_amd.protocol.dataplane import DataPlane
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.runtimes.sklearnserver import SKLearnModel

    model = SKLearnModel(name, model_dir)
    model.load()
    repo = ModelRepository()
    repo.update(model)
    return name, TestClient(create_app(DataPlane(repo)))


def test_isvc_to_prediction(iris_dir):
    server = FakeAPIServer()
    isvc_ctrl = InferenceServiceController(server).build()
    dep_ctrl = FakeDeploymentController(server).build()
    server.create(make_isvc(name="iris-e2e"))
    converge(isvc_ctrl, dep_ctrl)

    dep = server.get("apps/v1/Deployment", "default", "iris-e2e-predictor")
    # pod contract: storage-initializer feeds /mnt/models (simulated by
    # handing the booted server the downloaded dir)
    init = dep["spec"]["template"]["spec"]["initContainers"][0]
    assert init["args"] == ["s3://models/iris", "/mnt/models"]
    name, client = boot_rendered_server(dep, iris_dir)
    assert name == "iris-e2e"  # {{.Name}} templating reached the args

    # V2 infer through the same route the HTTPRoute/Ingress would carry
    r = client.post(
        f"/v2/models/{name}/infer",
        json={
            "inputs": [
                {
                    "name": "input-0",
                    "shape": [2, 4],
                    "datatype": "FP64",
                    "data": [5.1, 3.5, 1.4, 0.2, 6.7, 3.0, 5.2, 2.3],
                }
            ]
        },
    )
    assert r.status_code == 200
    assert r.json()["outputs"][0]["data"] == [0, 2]


def test_graph_cr_to_routed_prediction(iris_dir):
    """InferenceGraph CR → live controller → --graph-json → the ACTUAL
    router executing the graph against in-process predictors."""
    import asyncio

    import httpx

    server = FakeAPIServer()
    ig_ctrl = InferenceGraphController(server).build()
    dep_ctrl = FakeDeploymentController(server).build()
    server.create(
        {
            "apiVersion": "serving.kserve.io/v1alpha1",
            "kind": "InferenceGraph",
            "metadata": {"name": "iris-seq", "namespace": "default"},
            "spec": {
                "nodes": {
                    "root": {
                        "routerType": "Sequence",
                        "steps": [
                            {
                                "serviceUrl": "http://iris-svc/v1/models/iris:predict",
                                "data": "$request",
                            }
                        ],
                    }
                }
            },
        }
    )
    converge(ig_ctrl, dep_ctrl)
    d = server.get("apps/v1/Deployment", "default", "iris-seq")
    args = d["spec"]["template"]["spec"]["containers"][0]["args"]
    graph_spec = json.loads(args[args.index("--graph-json") + 1])

    # boot the predictor the graph targets
    from kserve_amd.graph.router import GraphRouter
    from kserve_amd.graph.types import InferenceGraphSpec
    from kserve_amd.model_repository import ModelRepository
    from kserve_amd.protocol.dataplane import DataPlane
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.runtimes.sklearnserver import SKLearnModel

    model = SKLearnModel("iris", iris_dir)
    model.load()
    repo = ModelRepository()
    repo.update(model)
    predictor_app = create_app(DataPlane(repo))
    transport = httpx.ASGITransport(app=predictor_app)

    router = GraphRouter(
        InferenceGraphSpec.from_dict(graph_spec), transport=transport
    )

    async def run():
        return await router.handle(
            {"instances": [[5.1, 3.5, 1.4, 0.2]]}, {}
        )

    code, body = asyncio.run(run())
    assert code == 200
    assert body["predictions"] == [0]
