"""Live InferenceGraph + TrainedModel controllers on the fake API server
(reference inferencegraph/controller.go + trainedmodel controller envtest
suites)."""

import json

from kserve_amd.controlplane.apiserver import FakeAPIServer
from kserve_amd.controlplane.controller import get_condition
from kserve_amd.controlplane.crd_controllers import (
    IG_GVK,
    TM_GVK,
    InferenceGraphController,
    TrainedModelController,
)
from kserve_amd.controlplane.isvc_controller import FakeDeploymentController
from tests.test_live_controller import converge


def make_graph(name="pipeline"):
    return {
        "apiVersion": "serving.kserve.io/v1alpha1",
        "kind": "InferenceGraph",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"serviceName": "tokenizer", "data": "$request"},
                        {"serviceName": "llm", "data": "$response"},
                    ],
                }
            }
        },
    }


class TestInferenceGraphController:
    def test_graph_deploys_router_and_reports_ready(self):
        server = FakeAPIServer()
        ig = InferenceGraphController(server).build()
        dep = FakeDeploymentController(server).build()
        server.create(make_graph())
        converge(ig, dep)
        d = server.get("apps/v1/Deployment", "default", "pipeline")
        args = d["spec"]["template"]["spec"]["containers"][0]["args"]
        assert args[0] == "--graph-json"
        spec = json.loads(args[1])
        assert spec["nodes"]["root"]["routerType"] == "Sequence"
        assert server.get("v1/Service", "default", "pipeline")
        cr = server.get(IG_GVK, "default", "pipeline")
        assert get_condition(cr["status"], "Ready")["status"] == "True"
        assert cr["status"]["url"].endswith("pipeline.default.svc.cluster.local")

    def test_graph_delete_cascades(self):
        server = FakeAPIServer()
        ig = InferenceGraphController(server).build()
        dep = FakeDeploymentController(server).build()
        server.create(make_graph())
        converge(ig, dep)
        server.delete(IG_GVK, "default", "pipeline")
        converge(ig, dep)
        assert server.try_get("apps/v1/Deployment", "default", "pipeline") is None


def make_tm(name, isvc="multi", uri="s3://b/m1"):
    return {
        "apiVersion": "serving.kserve.io/v1alpha1",
        "kind": "TrainedModel",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {
            "inferenceService": isvc,
            "model": {"storageUri": uri, "framework": "sklearn",
                      "memory": "1Gi"},
        },
    }


class TestTrainedModelController:
    def test_upsert_into_modelconfig(self):
        server = FakeAPIServer()
        tm = TrainedModelController(server).build()
        server.create(make_tm("m-a", uri="s3://b/a"))
        server.create(make_tm("m-b", uri="s3://b/b"))
        converge(tm)
        cm = server.get("v1/ConfigMap", "default", "modelconfig-multi-0")
        models = json.loads(cm["data"]["models.json"])
        assert [m["modelName"] for m in models] == ["m-a", "m-b"]
        assert models[0]["modelSpec"]["storageUri"] == "s3://b/a"
        cr = server.get(TM_GVK, "default", "m-a")
        assert get_condition(cr["status"], "Ready")["status"] == "True"
        assert "trainedmodel.finalizers" in cr["metadata"]["finalizers"]

    def test_update_replaces_entry(self):
        server = FakeAPIServer()
        tm = TrainedModelController(server).build()
        server.create(make_tm("m-a", uri="s3://b/v1"))
        converge(tm)
        cr = server.get(TM_GVK, "default", "m-a")
        cr["spec"]["model"]["storageUri"] = "s3://b/v2"
        server.update(cr)
        converge(tm)
        cm = server.get("v1/ConfigMap", "default", "modelconfig-multi-0")
        models = json.loads(cm["data"]["models.json"])
        assert len(models) == 1
        assert models[0]["modelSpec"]["storageUri"] == "s3://b/v2"

    def test_delete_removes_entry_via_finalizer(self):
        server = FakeAPIServer()
        tm = TrainedModelController(server).build()
        server.create(make_tm("m-a"))
        server.create(make_tm("m-b", uri="s3://b/b"))
        converge(tm)
        server.delete(TM_GVK, "default", "m-a")
        converge(tm)
        assert server.try_get(TM_GVK, "default", "m-a") is None
        cm = server.get("v1/ConfigMap", "default", "modelconfig-multi-0")
        models = json.loads(cm["data"]["models.json"])
        assert [m["modelName"] for m in models] == ["m-b"]

    def test_invalid_spec_condition(self):
        server = FakeAPIServer()
        tm = TrainedModelController(server).build()
        bad = make_tm("m-bad")
        del bad["spec"]["model"]["storageUri"]
        server.create(bad)
        converge(tm)
        cr = server.get(TM_GVK, "default", "m-bad")
        ready = get_condition(cr["status"], "Ready")
        assert ready["status"] == "False" and ready["reason"] == "InvalidSpec"


def test_manager_runs_all_controllers_threaded():
    """cmd/manager equivalent: one process, every controller on live
    threads, converging CRs end-to-end (threaded mode, not the test
    drive loop)."""
    import time

    from kserve_amd.controlplane.apiserver import FakeAPIServer
    from kserve_amd.controlplane.manager import build_controllers
    from kserve_amd.controlplane.isvc_controller import (
        FakeDeploymentController,
    )
    from tests.test_live_controller import make_isvc

    server = FakeAPIServer()
    controllers = build_controllers(server)
    dep_ctrl = FakeDeploymentController(server).build()
    controllers.append(dep_ctrl)
    for c in controllers:
        c.start()
    try:
        server.create(make_isvc(name="mgr-iris"))
        server.create(make_tm("mgr-tm", isvc="mgr-iris", uri="s3://b/m"))
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            isvc = server.try_get(
                "serving.kserve.io/v1beta1/InferenceService", "default",
                "mgr-iris",
            )
            cm = server.try_get(
                "v1/ConfigMap", "default", "modelconfig-mgr-iris-0"
            )
            if (
                isvc
                and get_condition(isvc.get("status", {}), "Ready")
                and get_condition(isvc["status"], "Ready")["status"] == "True"
                and cm is not None
            ):
                break
            time.sleep(0.05)
        isvc = server.get(
            "serving.kserve.io/v1beta1/InferenceService", "default", "mgr-iris"
        )
        assert get_condition(isvc["status"], "Ready")["status"] == "True"
        assert server.get("v1/ConfigMap", "default", "modelconfig-mgr-iris-0")
    finally:
        for c in controllers:
            c.stop()
