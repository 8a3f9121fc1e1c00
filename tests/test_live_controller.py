"""Live control-plane tests: controllers converging real(ish) state on the
fake API server — the envtest style of the reference
(rawkube_controller_test.go, canary_controller_test.go): create a CR, drive
the manager, assert the created objects and the status conditions; mutate
and delete and assert convergence again."""

import json

import pytest

from kserve_amd.controlplane.apiserver import (
    AlreadyExists,
    Conflict,
    FakeAPIServer,
    NotFound,
)
from kserve_amd.controlplane.controller import (
    create_or_update,
    get_condition,
)
from kserve_amd.controlplane.isvc_controller import (
    ISVC_GVK,
    FakeDeploymentController,
    InferenceServiceController,
)

DEP = "apps/v1/Deployment"
SVC = "v1/Service"


def make_isvc(name="iris", namespace="default", **annotations):
    return {
        "apiVersion": "serving.kserve.io/v1beta1",
        "kind": "InferenceService",
        "metadata": {
            "name": name,
            "namespace": namespace,
            "annotations": dict(annotations),
        },
        "spec": {
            "predictor": {
                "model": {
                    "modelFormat": {"name": "sklearn"},
                    "storageUri": "s3://models/iris",
                    "protocolVersion": "v2",
                }
            }
        },
    }


def converge(*controllers, seconds=5.0):
    """Interleaved manager loop: pump + process every controller until all
    queues (including delayed requeues) drain or the deadline passes."""
    import time

    deadline = time.monotonic() + seconds
    idle = 0
    while time.monotonic() < deadline:
        moved = False
        for c in controllers:
            moved |= c.pump_events(0.0) > 0
            while c.process_one(timeout=0.0):
                moved = True
        if moved:
            idle = 0
            continue
        if all(c.queue.empty() for c in controllers):
            idle += 1
            if idle >= 3:
                return
        time.sleep(0.01)


@pytest.fixture
def env():
    server = FakeAPIServer()
    isvc_ctrl = InferenceServiceController(server).build()
    dep_ctrl = FakeDeploymentController(server).build()
    return server, isvc_ctrl, dep_ctrl


# ---- fake API server semantics ---------------------------------------------

class TestFakeAPIServer:
    def test_create_get_conflict(self):
        s = FakeAPIServer()
        obj = s.create(make_isvc())
        assert obj["metadata"]["uid"]
        assert obj["metadata"]["generation"] == 1
        with pytest.raises(AlreadyExists):
            s.create(make_isvc())

    def test_update_optimistic_concurrency(self):
        s = FakeAPIServer()
        a = s.create(make_isvc())
        b = s.get(ISVC_GVK, "default", "iris")
        a["spec"]["predictor"]["minReplicas"] = 2
        s.update(a)
        b["spec"]["predictor"]["minReplicas"] = 3
        with pytest.raises(Conflict):
            s.update(b)  # stale resourceVersion

    def test_generation_bumps_on_spec_not_status(self):
        s = FakeAPIServer()
        obj = s.create(make_isvc())
        obj["status"] = {"conditions": [{"type": "Ready", "status": "True"}]}
        s.update_status(obj)
        cur = s.get(ISVC_GVK, "default", "iris")
        assert cur["metadata"]["generation"] == 1
        cur["spec"]["predictor"]["minReplicas"] = 5
        s.update(cur)
        assert s.get(ISVC_GVK, "default", "iris")["metadata"]["generation"] == 2

    def test_finalizer_blocks_delete_until_removed(self):
        s = FakeAPIServer()
        obj = s.create(make_isvc())
        obj["metadata"]["finalizers"] = ["x"]
        obj = s.update(obj)
        s.delete(ISVC_GVK, "default", "iris")
        cur = s.get(ISVC_GVK, "default", "iris")  # still there
        assert cur["metadata"]["deletionTimestamp"]
        cur["metadata"]["finalizers"] = []
        s.update(cur)
        with pytest.raises(NotFound):
            s.get(ISVC_GVK, "default", "iris")

    def test_owner_reference_cascade(self):
        s = FakeAPIServer()
        owner = s.create(make_isvc())
        child = {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "metadata": {"name": "c", "namespace": "default"},
            "spec": {},
        }
        create_or_update(s, child, owner=owner)
        assert s.try_get(DEP, "default", "c")
        s.delete(ISVC_GVK, "default", "iris")
        assert s.try_get(DEP, "default", "c") is None

    def test_watch_streams_events(self):
        s = FakeAPIServer()
        w = s.watch(ISVC_GVK)
        s.create(make_isvc())
        ev = w.next(timeout=1)
        assert ev.type == "ADDED" and ev.object["metadata"]["name"] == "iris"

    def test_create_or_update_semantic_noop(self):
        s = FakeAPIServer()
        d = {
            "apiVersion": "v1",
            "kind": "Service",
            "metadata": {"name": "a", "namespace": "default"},
            "spec": {"selector": {"app": "a"}},
        }
        first = create_or_update(s, d)
        rv = first["metadata"]["resourceVersion"]
        second = create_or_update(s, d)
        assert second["metadata"]["resourceVersion"] == rv  # no-op elided


# ---- live InferenceService controller ---------------------------------------

class TestLiveISVCController:
    def test_isvc_converges_to_ready(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(make_isvc())
        converge(isvc_ctrl, dep_ctrl)
        dep = server.get(DEP, "default", "iris-predictor")
        assert dep["status"]["availableReplicas"] == 1
        assert server.get(SVC, "default", "iris-predictor")
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert get_condition(isvc["status"], "PredictorReady")["status"] == "True"
        assert get_condition(isvc["status"], "Ready")["status"] == "True"
        assert isvc["status"]["url"].startswith("http://iris-default.")
        assert "inferenceservice.finalizers" in isvc["metadata"]["finalizers"]

    def test_oci_image_volume_advisory_condition(self, env):
        """oci:// models get the non-blocking advisory condition
        (reference controller.go:774-831): ModelcarFallback by default,
        flipping when the cluster advertises ImageVolume support."""
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc(name="ocimodel")
        obj["spec"]["predictor"]["model"]["storageUri"] = "oci://reg/m:1"
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        isvc = server.get(ISVC_GVK, "default", "ocimodel")
        adv = get_condition(isvc["status"], "OCIImageVolumeAdvisory")
        assert adv["status"] == "False" and adv["reason"] == "ModelcarFallback"
        # s3 models carry no advisory
        server.create(make_isvc(name="plain"))
        converge(isvc_ctrl, dep_ctrl)
        plain = server.get(ISVC_GVK, "default", "plain")
        assert get_condition(plain["status"], "OCIImageVolumeAdvisory") is None
        # flip the cluster capability through the config system
        server.create({
            "apiVersion": "v1", "kind": "ConfigMap",
            "metadata": {"name": "inferenceservice-config",
                         "namespace": "kserve"},
            "data": {"deploy": '{"imageVolumeAvailable": true}'},
        })
        isvc_ctrl.queue.add(("default", "ocimodel"))
        converge(isvc_ctrl, dep_ctrl)
        isvc = server.get(ISVC_GVK, "default", "ocimodel")
        adv = get_condition(isvc["status"], "OCIImageVolumeAdvisory")
        assert adv["status"] == "True"
        assert adv["reason"] == "ImageVolumeAvailable"

    def test_requeues_until_pods_ready(self, env):
        server, isvc_ctrl, _ = env
        # deployment controller that stays unavailable for a few rounds
        slow = FakeDeploymentController(server, delay_updates=3).build()
        server.create(make_isvc())
        converge(isvc_ctrl, slow)
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert get_condition(isvc["status"], "Ready")["status"] == "True"

    def test_canary_rollout_and_promotion(self, env):
        """Reference canary e2e (test_canary*.py): setting
        canaryTrafficPercent creates the -canary Deployment; promotion
        (removing the percent) prunes it again."""
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc(name="cnr")
        obj["spec"]["predictor"]["canaryTrafficPercent"] = 10
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "cnr-predictor")
        assert server.try_get(DEP, "default", "cnr-predictor-canary")
        # promotion: drop the canary percent
        cur = server.get(ISVC_GVK, "default", "cnr")
        del cur["spec"]["predictor"]["canaryTrafficPercent"]
        server.update(cur)
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "cnr-predictor")
        assert server.try_get(DEP, "default", "cnr-predictor-canary") is None
        isvc = server.get(ISVC_GVK, "default", "cnr")
        assert get_condition(isvc["status"], "Ready")["status"] == "True"

    def test_drift_is_repaired(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(make_isvc())
        converge(isvc_ctrl, dep_ctrl)
        dep = server.get(DEP, "default", "iris-predictor")
        want_image = dep["spec"]["template"]["spec"]["containers"][0]["image"]
        dep["spec"]["template"]["spec"]["containers"][0]["image"] = "evil:1"
        server.update(dep)
        converge(isvc_ctrl, dep_ctrl)
        dep = server.get(DEP, "default", "iris-predictor")
        assert (
            dep["spec"]["template"]["spec"]["containers"][0]["image"]
            == want_image
        )

    def test_delete_runs_finalizer_and_cascades(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(make_isvc())
        converge(isvc_ctrl, dep_ctrl)
        # external resource the finalizer must clean up
        server.create(
            {
                "apiVersion": "v1",
                "kind": "ConfigMap",
                "metadata": {"name": "modelconfig-iris-0", "namespace": "default"},
                "data": {},
            }
        )
        server.delete(ISVC_GVK, "default", "iris")
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(ISVC_GVK, "default", "iris") is None
        assert server.try_get("v1/ConfigMap", "default", "modelconfig-iris-0") is None
        assert server.try_get(DEP, "default", "iris-predictor") is None  # GC

    def test_canary_pair_then_promote(self, env):
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc()
        obj["spec"]["predictor"]["canaryTrafficPercent"] = 20
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "iris-predictor-canary")
        assert server.try_get(SVC, "default", "iris-predictor-canary")
        # promote: clear the canary percent -> canary objects pruned
        cur = server.get(ISVC_GVK, "default", "iris")
        del cur["spec"]["predictor"]["canaryTrafficPercent"]
        server.update(cur)
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "iris-predictor-canary") is None
        assert server.try_get(SVC, "default", "iris-predictor-canary") is None

    def test_hpa_and_keda_switch(self, env):
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc()
        obj["spec"]["predictor"]["minReplicas"] = 1
        obj["spec"]["predictor"]["maxReplicas"] = 5
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        hpa = server.get(
            "autoscaling/v2/HorizontalPodAutoscaler", "default", "iris-predictor"
        )
        assert hpa["spec"]["maxReplicas"] == 5
        # switch to KEDA via annotation -> ScaledObject replaces HPA
        cur = server.get(ISVC_GVK, "default", "iris")
        cur["metadata"]["annotations"][
            "serving.kserve.io/autoscalerClass"
        ] = "keda"
        server.update(cur)
        converge(isvc_ctrl, dep_ctrl)
        assert (
            server.try_get(
                "autoscaling/v2/HorizontalPodAutoscaler",
                "default",
                "iris-predictor",
            )
            is None
        )
        so = server.get("keda.sh/v1alpha1/ScaledObject", "default", "iris-predictor")
        assert so["spec"]["maxReplicaCount"] == 5

    def test_invalid_spec_sets_condition(self, env):
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc()
        obj["spec"]["predictor"]["model"]["storageUri"] = "ftp://nope/x"
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        isvc = server.get(ISVC_GVK, "default", "iris")
        ready = get_condition(isvc["status"], "Ready")
        assert ready["status"] == "False"
        assert ready["reason"] == "InvalidSpec"
        assert server.try_get(DEP, "default", "iris-predictor") is None

    def test_transformer_deployment_with_predictor_host(self, env):
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc()
        obj["spec"]["transformer"] = {
            "containers": [
                {"name": "kserve-container", "image": "my-transformer:1"}
            ]
        }
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        dep = server.get(DEP, "default", "iris-transformer")
        args = dep["spec"]["template"]["spec"]["containers"][0]["args"]
        assert "--predictor_host" in args
        assert "iris-predictor.default" in args
        # ingress should target the transformer
        ing = server.get("networking.k8s.io/v1/Ingress", "default", "iris")
        backend = ing["spec"]["rules"][0]["http"]["paths"][0]["backend"]
        assert backend["service"]["name"] == "iris-transformer"
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert (
            get_condition(isvc["status"], "TransformerReady")["status"] == "True"
        )

    def test_stop_annotation_tears_down(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(make_isvc())
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "iris-predictor")
        cur = server.get(ISVC_GVK, "default", "iris")
        cur["metadata"]["annotations"]["serving.kserve.io/stop"] = "true"
        server.update(cur)
        converge(isvc_ctrl, dep_ctrl)
        assert server.try_get(DEP, "default", "iris-predictor") is None
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert get_condition(isvc["status"], "Stopped")["status"] == "True"
        assert get_condition(isvc["status"], "Ready")["status"] == "False"

    def test_configmap_drives_domain_and_gateway_api(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(
            {
                "apiVersion": "v1",
                "kind": "ConfigMap",
                "metadata": {
                    "name": "inferenceservice-config",
                    "namespace": "kserve",
                },
                "data": {
                    "ingress": json.dumps(
                        {
                            "ingressDomain": "models.corp",
                            "enableGatewayApi": True,
                        }
                    )
                },
            }
        )
        server.create(make_isvc())
        converge(isvc_ctrl, dep_ctrl)
        # HTTPRoute backend instead of Ingress; domain from config
        route = server.get(
            "gateway.networking.k8s.io/v1/HTTPRoute", "default", "iris"
        )
        assert route["spec"]["hostnames"] == ["iris-default.models.corp"]
        assert server.try_get("networking.k8s.io/v1/Ingress", "default", "iris") is None
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert isvc["status"]["url"] == "http://iris-default.models.corp"

    def test_servingruntime_cr_is_used(self, env):
        server, isvc_ctrl, dep_ctrl = env
        server.create(
            {
                "apiVersion": "serving.kserve.io/v1alpha1",
                "kind": "ServingRuntime",
                "metadata": {"name": "my-rt", "namespace": "default"},
                "spec": {
                    "supportedModelFormats": [
                        {"name": "customfmt", "autoSelect": True}
                    ],
                    "protocolVersions": ["v2"],
                    "containers": [
                        {
                            "name": "kserve-container",
                            "image": "custom-rt:9",
                            "args": ["--model_dir=/mnt/models"],
                        }
                    ],
                },
            }
        )
        obj = make_isvc()
        obj["spec"]["predictor"]["model"]["modelFormat"]["name"] = "customfmt"
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        dep = server.get(DEP, "default", "iris-predictor")
        img = dep["spec"]["template"]["spec"]["containers"][0]["image"]
        assert img == "custom-rt:9"

    def test_serverless_mode_creates_knative_service(self, env):
        server, isvc_ctrl, dep_ctrl = env
        obj = make_isvc(**{"serving.kserve.io/deploymentMode": "Serverless"})
        server.create(obj)
        converge(isvc_ctrl, dep_ctrl)
        ksvc = server.get(
            "serving.knative.dev/v1/Service", "default", "iris-predictor"
        )
        assert ksvc["spec"]["traffic"][0]["percent"] == 100
        # not ready until the (simulated) knative controller reports it
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert get_condition(isvc["status"], "Ready")["status"] == "False"
        ksvc["status"] = {
            "conditions": [{"type": "Ready", "status": "True"}]
        }
        server.update_status(ksvc)
        converge(isvc_ctrl, dep_ctrl)
        isvc = server.get(ISVC_GVK, "default", "iris")
        assert get_condition(isvc["status"], "Ready")["status"] == "True"


def test_podmetrics_scale_metric_deploys_otel_collector(env):
    """Custom (PodMetrics) scale metrics deploy the OTel collector sidecar
    CR feeding the scaler (reference otel_reconciler.go wiring)."""
    server, isvc_ctrl, dep_ctrl = env
    obj = make_isvc()
    obj["spec"]["predictor"]["minReplicas"] = 1
    obj["spec"]["predictor"]["maxReplicas"] = 4
    obj["spec"]["predictor"]["scaleMetric"] = "llm_tokens_per_second"
    server.create(obj)
    converge(isvc_ctrl, dep_ctrl)
    otel = server.get(
        "opentelemetry.io/v1beta1/OpenTelemetryCollector",
        "default",
        "iris-predictor",
    )
    cfg = otel["spec"]["config"]
    inc = cfg["processors"]["filter/metrics"]["metrics"]["include"]
    assert inc["metric_names"] == ["llm_tokens_per_second"]
    assert otel["spec"]["mode"] == "sidecar"
    # switch back to cpu -> collector pruned
    cur = server.get(ISVC_GVK, "default", "iris")
    cur["spec"]["predictor"]["scaleMetric"] = "cpu"
    server.update(cur)
    converge(isvc_ctrl, dep_ctrl)
    assert (
        server.try_get(
            "opentelemetry.io/v1beta1/OpenTelemetryCollector",
            "default",
            "iris-predictor",
        )
        is None
    )


def test_cluster_storage_container_overrides_initializer(env):
    """ClusterStorageContainer CRs replace the storage-initializer image
    for matching URI formats (reference
    storage_initializer_injector.go:123-199)."""
    server, isvc_ctrl, dep_ctrl = env
    server.create(
        {
            "apiVersion": "serving.kserve.io/v1alpha1",
            "kind": "ClusterStorageContainer",
            "metadata": {"name": "custom-s3"},
            "spec": {
                "supportedUriFormats": [{"prefix": "s3://models/"}],
                "container": {
                    "name": "storage-initializer",
                    "image": "corp/custom-s3-init:2",
                    "env": [{"name": "S3_USE_ACCEL", "value": "1"}],
                },
            },
        }
    )
    server.create(make_isvc())  # storageUri s3://models/iris
    converge(isvc_ctrl, dep_ctrl)
    dep = server.get(DEP, "default", "iris-predictor")
    init = dep["spec"]["template"]["spec"]["initContainers"][0]
    assert init["image"] == "corp/custom-s3-init:2"
    assert {"name": "S3_USE_ACCEL", "value": "1"} in init["env"]
    # args keep the (uri, dest) contract
    assert init["args"][0] == "s3://models/iris"
