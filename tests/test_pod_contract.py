"""SURVEY.md §7.1 pod / data-plane contract conformance.

These interfaces are the stable boundary between control plane and data
plane; the reference's Go layer, presets, and clients all assume them
bit-for-bit (pkg/constants/constants.go, kserve_storage.py:62,
cmd/agent/main.go). This test pins OUR constants and rendered manifests
to those exact values so a refactor cannot silently break pod specs or
wire compatibility."""

from kserve_amd import constants
from kserve_amd.controlplane import webhook


class TestNamesPathsPorts:
    def test_container_names(self):
        assert constants.INFERENCE_CONTAINER == "kserve-container"
        assert constants.STORAGE_INITIALIZER_CONTAINER == "storage-initializer"
        assert constants.TRANSFORMER_CONTAINER == "transformer-container"
        assert constants.WORKER_CONTAINER == "worker-container"
        assert constants.AGENT_CONTAINER == "agent"

    def test_paths(self):
        assert constants.MODEL_MOUNT_PATH == "/mnt/models"
        assert constants.MODEL_CONFIG_MOUNT_PATH == "/mnt/configs"
        assert constants.PVC_MOUNT_PATH == "/mnt/pvc"

    def test_ports(self):
        assert constants.HTTP_PORT == 8080
        assert constants.GRPC_PORT == 8081
        assert constants.AGENT_PORT == 9081
        assert constants.LOG_MARSHALLER_PORT == 9083
        assert constants.ROUTER_PORT == 8080

    def test_binary_extension_header(self):
        assert (constants.INFERENCE_CONTENT_LENGTH_HEADER
                == "inference-content-length")

    def test_internal_annotations(self):
        assert webhook.ANN_STORAGE_URI == (
            "internal.serving.kserve.io/storage-initializer-sourceuri")
        assert webhook.ANN_LOGGER == "internal.serving.kserve.io/logger"
        assert webhook.ANN_BATCHER == "internal.serving.kserve.io/batcher"
        assert webhook.ANN_AGENT == "internal.serving.kserve.io/agent"


class TestRenderedPodContract:
    """The mutator must produce exactly the contract shapes."""

    def test_storage_initializer_contract(self):
        pod = {
            "metadata": {"annotations": {
                webhook.ANN_STORAGE_URI: "s3://bucket/model"}},
            "spec": {"containers": [
                {"name": "kserve-container", "image": "x"}]},
        }
        out = webhook.mutate_pod(pod)
        init = out["spec"]["initContainers"][0]
        assert init["name"] == "storage-initializer"
        assert init["args"] == ["s3://bucket/model", "/mnt/models"]
        mount = init["volumeMounts"][0]
        assert mount["mountPath"] == "/mnt/models"
        # model container sees the same volume read-only
        kc = out["spec"]["containers"][0]
        m = [v for v in kc["volumeMounts"]
             if v["mountPath"] == "/mnt/models"][0]
        assert m["readOnly"] is True

    def test_agent_proxy_port_contract(self):
        pod = {
            "metadata": {"annotations": {webhook.ANN_AGENT: "true"}},
            "spec": {"containers": [
                {"name": "kserve-container", "image": "x"}]},
        }
        out = webhook.mutate_pod(pod)
        agent = [c for c in out["spec"]["containers"]
                 if c["name"] == "agent"][0]
        assert {"containerPort": 9081, "name": "agent-port"} in agent["ports"]
        assert "--port" in agent["args"]
        assert "9081" in agent["args"]

    def test_pvc_fast_path_mounts_at_mnt_pvc(self):
        pod = {
            "metadata": {"annotations": {
                webhook.ANN_STORAGE_URI: "pvc://claim/models/a"}},
            "spec": {"containers": [
                {"name": "kserve-container", "image": "x"}]},
        }
        out = webhook.mutate_pod(pod)
        init = out["spec"]["initContainers"][0]
        pvc_mount = [m for m in init["volumeMounts"]
                     if m["name"] == "kserve-pvc-source"][0]
        assert pvc_mount["mountPath"] == "/mnt/pvc"
        assert pvc_mount["readOnly"] is True
