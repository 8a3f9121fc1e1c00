"""LocalModel cluster orchestration: cache CR → node CRs → PV/PVC + download
Jobs → aggregated status (reference localmodelcache_reconciler.go +
localmodelnode/controller.go envtest suites)."""

from kserve_amd.controlplane.apiserver import FakeAPIServer
from kserve_amd.controlplane.localmodel_controllers import (
    CACHE_GVK,
    JOB_GVK,
    NODE_CR_GVK,
    PV_GVK,
    PVC_GVK,
    FakeJobController,
    LocalModelCacheController,
    LocalModelNodeController,
)
from tests.test_live_controller import converge


def make_env(nodes=("gpu-node-1", "gpu-node-2"), fail_jobs=None):
    server = FakeAPIServer()
    server.create(
        {
            "apiVersion": "serving.kserve.io/v1alpha1",
            "kind": "LocalModelNodeGroup",
            "metadata": {"name": "gpu-group"},
            "spec": {
                "nodeSelector": {"group": "gpu"},
                "persistentVolumeSpec": {
                    "hostPath": {"path": "/models/gpu-group"},
                    "capacity": {"storage": "200Gi"},
                },
            },
        }
    )
    for n in nodes:
        server.create(
            {
                "apiVersion": "v1",
                "kind": "Node",
                "metadata": {"name": n, "labels": {"group": "gpu"}},
            }
        )
    cache_ctrl = LocalModelCacheController(server).build()
    node_ctrls = [
        LocalModelNodeController(server, n).build() for n in nodes
    ]
    job_ctrl = FakeJobController(server, fail_names=fail_jobs).build()
    return server, [cache_ctrl, *node_ctrls, job_ctrl]


def make_cache(name="llama-8b", uri="hf://meta/llama-3-8b"):
    return {
        "apiVersion": "serving.kserve.io/v1alpha1",
        "kind": "LocalModelCache",
        "metadata": {"name": name},
        "spec": {
            "sourceModelUri": uri,
            "modelSize": "16Gi",
            "nodeGroups": ["gpu-group"],
        },
    }


def test_cache_fans_out_to_nodes_and_pv_pvc():
    server, ctrls = make_env()
    server.create(make_cache())
    converge(*ctrls)
    # per-node CRs carry the model entry
    for n in ("gpu-node-1", "gpu-node-2"):
        cr = server.get(NODE_CR_GVK, "", n)
        models = cr["spec"]["localModels"]
        assert models[0]["modelName"] == "llama-8b"
        assert models[0]["sourceModelUri"] == "hf://meta/llama-3-8b"
    # PV from the node group's template + bound PVC in the jobs namespace
    pv = server.get(PV_GVK, "", "llama-8b-gpu-group-pv")
    assert pv["spec"]["hostPath"]["path"] == "/models/gpu-group"
    assert pv["spec"]["capacity"]["storage"] == "200Gi"
    pvc = server.get(PVC_GVK, "kserve-localmodel-jobs", "llama-8b-gpu-group-pv")
    assert pvc["spec"]["volumeName"] == "llama-8b-gpu-group-pv"


def test_download_jobs_run_and_status_aggregates():
    server, ctrls = make_env()
    server.create(make_cache())
    converge(*ctrls)
    # a download Job per node, storage-initializer image, node pinned
    job = server.get(
        JOB_GVK, "kserve-localmodel-jobs", "llama-8b-gpu-node-1-download"
    )
    podspec = job["spec"]["template"]["spec"]
    assert podspec["nodeName"] == "gpu-node-1"
    assert podspec["containers"][0]["args"][0] == "hf://meta/llama-3-8b"
    # fake job controller marked them succeeded -> cache reports copies
    cache = server.get(CACHE_GVK, "", "llama-8b")
    assert cache["status"]["copies"] == {"total": 2, "available": 2}
    assert cache["status"]["nodeStatus"] == {
        "gpu-node-1": "NodeDownloaded",
        "gpu-node-2": "NodeDownloaded",
    }


def test_failed_download_is_reported():
    server, ctrls = make_env(
        nodes=("gpu-node-1",),
        fail_jobs={"llama-8b-gpu-node-1-download"},
    )
    server.create(make_cache())
    converge(*ctrls)
    cache = server.get(CACHE_GVK, "", "llama-8b")
    assert cache["status"]["nodeStatus"]["gpu-node-1"] == "NodeDownloadError"
    assert cache["status"]["copies"]["available"] == 0


def test_cache_delete_cleans_node_entries():
    server, ctrls = make_env(nodes=("gpu-node-1",))
    server.create(make_cache())
    converge(*ctrls)
    assert server.get(NODE_CR_GVK, "", "gpu-node-1")["spec"]["localModels"]
    server.delete(CACHE_GVK, "", "llama-8b")
    converge(*ctrls)
    cr = server.try_get(NODE_CR_GVK, "", "gpu-node-1")
    # entry removed (or whole CR GC'd via ownerReference)
    assert cr is None or all(
        m["modelName"] != "llama-8b" for m in cr["spec"].get("localModels", [])
    )
    # node controller drops the Job for the removed model
    assert (
        server.try_get(
            JOB_GVK, "kserve-localmodel-jobs", "llama-8b-gpu-node-1-download"
        )
        is None
    )


def test_two_caches_share_a_node_cr():
    server, ctrls = make_env(nodes=("gpu-node-1",))
    server.create(make_cache("m-a", "s3://b/a"))
    server.create(make_cache("m-b", "s3://b/b"))
    converge(*ctrls)
    cr = server.get(NODE_CR_GVK, "", "gpu-node-1")
    names = sorted(m["modelName"] for m in cr["spec"]["localModels"])
    assert names == ["m-a", "m-b"]
    st = cr["status"]["modelStatus"]
    assert st == {"m-a": "NodeDownloaded", "m-b": "NodeDownloaded"}
