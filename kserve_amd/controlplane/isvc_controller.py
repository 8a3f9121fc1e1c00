"""Live InferenceService controller: watch → reconcile → status, against a
pluggable API server.

Reference parity: pkg/controller/v1beta1/inferenceservice/controller.go —
Reconcile (:122-418): config load (:134-142), deployment-mode resolve (:155),
finalizer add/remove + deleteExternalResources (:176-213, :738-758),
component fan-out (:281-305), ingress (:360-401), modelconfig (:404-410),
updateStatus with semantic-equality short-circuit (:420-455); watches on
owned Deployments/Services propagate pod readiness back (:624-736);
PropagateModelStatus requeues every second until ready
(components/predictor.go:285-289); Stopped-condition teardown
(controller.go:306-338).

The desired manifests come from reconciler.desired_state (the round-1 pure
renderer); this module is the loop around it: apply with owner references
and semantic diff, prune orphans, set status conditions, requeue until the
Deployments report available.
"""

from __future__ import annotations

import copy
from typing import Dict, List, Optional, Tuple

from kserve_amd.controlplane.apiserver import NotFound, gvk_of
from kserve_amd.controlplane.configmap import load_config
from kserve_amd.controlplane.controller import (
    Controller,
    Result,
    create_or_update,
    delete_if_exists,
    get_condition,
    set_condition,
)
from kserve_amd.controlplane.reconciler import desired_state
from kserve_amd.controlplane.servingruntime import (
    ServingRuntime,
    SupportedModelFormat,
    default_cluster_runtimes,
)
from kserve_amd.controlplane.v1beta1 import (
    BatcherSpec,
    ExplainerSpec,
    FrameworkSpec,
    InferenceService,
    InferenceServiceSpec,
    LoggerSpec,
    ModelFormat,
    PredictorModelSpec,
    PredictorSpec,
    TransformerSpec,
    ValidationError,
    WorkerSpec,
)

ISVC_GVK = "serving.kserve.io/v1beta1/InferenceService"
SR_GVK = "serving.kserve.io/v1alpha1/ServingRuntime"
CSR_GVK = "serving.kserve.io/v1alpha1/ClusterServingRuntime"
CSC_GVK = "serving.kserve.io/v1alpha1/ClusterStorageContainer"
FINALIZER = "inferenceservice.finalizers"

# every kind the controller may create (for pruning + owned watches)
MANAGED_GVKS = (
    "apps/v1/Deployment",
    "v1/Service",
    "autoscaling/v2/HorizontalPodAutoscaler",
    "keda.sh/v1alpha1/ScaledObject",
    "networking.istio.io/v1beta1/VirtualService",
    "gateway.networking.k8s.io/v1/HTTPRoute",
    "networking.k8s.io/v1/Ingress",
    "serving.knative.dev/v1/Service",
    "opentelemetry.io/v1beta1/OpenTelemetryCollector",
)


# -- manifest <-> dataclass conversion ---------------------------------------

def _ext_fields(dst, src: Dict) -> None:
    """ComponentExtensionSpec fields shared by all components."""
    if "minReplicas" in src:
        dst.min_replicas = src["minReplicas"]
    if "maxReplicas" in src:
        dst.max_replicas = src["maxReplicas"]
    if "scaleTarget" in src:
        dst.scale_target = src["scaleTarget"]
    if "scaleMetric" in src:
        dst.scale_metric = src["scaleMetric"]
    if "canaryTrafficPercent" in src:
        dst.canary_traffic_percent = src["canaryTrafficPercent"]
    if "timeout" in src:
        dst.timeout_seconds = src["timeout"]
    if src.get("logger"):
        lg = src["logger"]
        dst.logger = LoggerSpec(mode=lg.get("mode", "all"), url=lg.get("url"))
    if src.get("batcher"):
        b = src["batcher"]
        dst.batcher = BatcherSpec(
            max_batch_size=b.get("maxBatchSize", 32),
            max_latency_ms=b.get("maxLatency", 5000),
        )
    if src.get("serviceAccountName"):
        dst.service_account_name = src["serviceAccountName"]


def _framework_spec(src: Dict) -> FrameworkSpec:
    return FrameworkSpec(
        storage_uri=src.get("storageUri"),
        runtime_version=src.get("runtimeVersion"),
        protocol_version=src.get("protocolVersion", "v1"),
        resources=src.get("resources", {}) or {},
    )


_FRAMEWORK_KEYS = (
    "sklearn", "xgboost", "lightgbm", "huggingface", "pmml", "paddle",
    "triton", "tensorflow", "pytorch", "onnx",
)


def isvc_from_manifest(obj: Dict) -> InferenceService:
    """Parse an InferenceService CR dict into the typed spec."""
    md = obj.get("metadata", {})
    spec = obj.get("spec", {}) or {}
    psrc = spec.get("predictor", {}) or {}
    p = PredictorSpec()
    _ext_fields(p, psrc)
    if psrc.get("model"):
        m = psrc["model"]
        fmt = m.get("modelFormat") or {}
        p.model = PredictorModelSpec(
            model_format=ModelFormat(
                name=fmt.get("name", ""), version=fmt.get("version")
            ),
            storage_uri=m.get("storageUri"),
            runtime=m.get("runtime"),
            protocol_version=m.get("protocolVersion", "v1"),
            resources=m.get("resources", {}) or {},
            args=m.get("args", []) or [],
            image=m.get("image"),
        )
    for fw in _FRAMEWORK_KEYS:
        if psrc.get(fw):
            setattr(p, fw, _framework_spec(psrc[fw]))
    if psrc.get("containers"):
        p.containers = copy.deepcopy(psrc["containers"])
    if psrc.get("workerSpec"):
        w = psrc["workerSpec"]
        p.worker = WorkerSpec(
            size=w.get("size", 1),
            pipeline_parallel_size=w.get("pipelineParallelSize"),
            tensor_parallel_size=w.get("tensorParallelSize"),
        )
    transformer = None
    if spec.get("transformer"):
        transformer = TransformerSpec(
            containers=copy.deepcopy(spec["transformer"].get("containers", []))
        )
        _ext_fields(transformer, spec["transformer"])
    explainer = None
    if spec.get("explainer"):
        esrc = spec["explainer"]
        explainer = ExplainerSpec(
            containers=copy.deepcopy(esrc.get("containers", []))
        )
        if esrc.get("art"):
            explainer.art = _framework_spec(esrc["art"])
        _ext_fields(explainer, esrc)
    return InferenceService(
        name=md["name"],
        namespace=md.get("namespace", "default"),
        spec=InferenceServiceSpec(
            predictor=p, transformer=transformer, explainer=explainer
        ),
        annotations=dict(md.get("annotations", {}) or {}),
        labels=dict(md.get("labels", {}) or {}),
    )


def runtime_from_manifest(obj: Dict) -> ServingRuntime:
    """ServingRuntime / ClusterServingRuntime CR -> catalog entry."""
    spec = obj.get("spec", {}) or {}
    formats = [
        SupportedModelFormat(
            name=f.get("name", ""),
            version=f.get("version"),
            auto_select=f.get("autoSelect", False),
            priority=f.get("priority", 1),
        )
        for f in spec.get("supportedModelFormats", []) or []
    ]
    containers = spec.get("containers", []) or []
    return ServingRuntime(
        name=obj["metadata"]["name"],
        supported_model_formats=formats,
        container=copy.deepcopy(containers[0]) if containers else {},
        protocol_versions=spec.get("protocolVersions", ["v1", "v2"]),
        disabled=spec.get("disabled", False),
        multi_model=spec.get("multiModel", False),
        cluster_scoped=obj.get("kind") == "ClusterServingRuntime",
        workers=bool(spec.get("workerSpec")),
    )


# -- the controller ----------------------------------------------------------

class InferenceServiceController:
    def __init__(
        self,
        server,
        config_namespace: str = "kserve",
        builtin_runtimes: bool = True,
    ):
        self.server = server
        self.config_namespace = config_namespace
        self.builtin_runtimes = builtin_runtimes

    # -- runtimes ----------------------------------------------------------
    def _load_runtimes(self, namespace: str) -> List[ServingRuntime]:
        ns_rts = [
            runtime_from_manifest(o)
            for o in self.server.list(SR_GVK, namespace)
        ]
        cl_rts = [
            runtime_from_manifest(o) for o in self.server.list(CSR_GVK)
        ]
        # creation-order sort mirrors GetSupportingRuntimes' tie-break
        if self.builtin_runtimes:
            cl_rts += default_cluster_runtimes()
        return ns_rts + cl_rts

    # -- external-resource cleanup (finalizer body) ------------------------
    def _delete_external_resources(self, obj: Dict) -> None:
        """reference deleteExternalResources (controller.go:738-758): drop
        the multi-model ConfigMap the ISVC owned."""
        md = obj["metadata"]
        delete_if_exists(
            self.server,
            "v1/ConfigMap",
            md.get("namespace", "default"),
            f"modelconfig-{md['name']}-0",
        )

    # -- status ------------------------------------------------------------
    def _deployment_ready(self, namespace: str, name: str) -> Tuple[bool, str]:
        dep = self.server.try_get("apps/v1/Deployment", namespace, name)
        if dep is None:
            return False, "DeploymentNotCreated"
        want = dep.get("spec", {}).get("replicas", 1)
        have = dep.get("status", {}).get("availableReplicas", 0)
        if have >= max(want, 1):
            return True, ""
        return False, "WaitingForPods"

    def _update_status(self, obj: Dict, status: Dict) -> None:
        """updateStatus equality short-circuit (controller.go:420-455)."""
        if obj.get("status", {}) == status:
            return
        newobj = copy.deepcopy(obj)
        newobj["status"] = status
        self.server.update_status(newobj)

    # -- reconcile ---------------------------------------------------------
    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        namespace, name = key
        obj = self.server.try_get(ISVC_GVK, namespace, name)
        if obj is None:
            return None  # deleted; GC cascaded owned objects
        md = obj["metadata"]

        # deletion: run finalizer, then release it
        if md.get("deletionTimestamp"):
            if FINALIZER in (md.get("finalizers") or []):
                self._delete_external_resources(obj)
                newobj = copy.deepcopy(obj)
                newobj["metadata"]["finalizers"] = [
                    f for f in md["finalizers"] if f != FINALIZER
                ]
                self.server.update(newobj)
            return None

        # ensure finalizer
        if FINALIZER not in (md.get("finalizers") or []):
            newobj = copy.deepcopy(obj)
            newobj["metadata"].setdefault("finalizers", []).append(FINALIZER)
            obj = self.server.update(newobj)
            md = obj["metadata"]

        cfg = load_config(self.server, self.config_namespace)

        # Stopped condition: the force-stop annotation tears workloads down
        # (reference controller.go:306-338)
        status = copy.deepcopy(obj.get("status", {}) or {})
        if (md.get("annotations") or {}).get(
            "serving.kserve.io/stop"
        ) == "true":
            self._prune(namespace, name, keep=set())
            set_condition(status, "Stopped", "True", reason="Stopped")
            set_condition(status, "Ready", "False", reason="Stopped")
            self._update_status(obj, status)
            return None

        try:
            isvc = isvc_from_manifest(obj)
            manifests = desired_state(isvc, self._load_runtimes(namespace), cfg)
        except (ValidationError, LookupError, ValueError) as e:
            set_condition(
                status, "Ready", "False",
                reason="InvalidSpec", message=str(e),
            )
            self._update_status(obj, status)
            return None  # invalid spec: wait for the user to fix it

        # credentials: ServiceAccount secrets -> storage-initializer env
        # (reference CreateSecretVolumeAndEnv via the pod webhook)
        sa_name = isvc.spec.predictor.service_account_name
        cred = None
        if sa_name:
            from kserve_amd.controlplane.credentials import (
                CredentialsBuilder,
                inject_credentials,
            )

            cred = CredentialsBuilder(self.server, namespace).for_service_account(
                sa_name
            )

        # custom storage containers: CSC CRs override the injected
        # storage-initializer by URI format
        csc_spec = None
        uri = isvc.spec.predictor.storage_uri
        if uri:
            from kserve_amd.controlplane.webhook import (
                apply_storage_container,
                resolve_storage_container,
            )

            csc_spec = resolve_storage_container(
                uri, self.server.list(CSC_GVK)
            )

        # apply all desired manifests with owner references
        applied_keys = set()
        for m in manifests:
            if cred and m.get("kind") == "Deployment":
                inject_credentials(m["spec"]["template"], *cred)
            if csc_spec and m.get("kind") == "Deployment":
                apply_storage_container(m["spec"]["template"], csc_spec)
            m["metadata"].setdefault("namespace", namespace)
            m["metadata"].setdefault("labels", {})[
                "serving.kserve.io/inferenceservice"
            ] = name
            applied = create_or_update(self.server, m, owner=obj)
            applied_keys.add(
                (gvk_of(applied), applied["metadata"]["name"])
            )
        self._prune(namespace, name, keep=applied_keys)

        # -- status conditions ------------------------------------------
        mode = (md.get("annotations") or {}).get(
            "serving.kserve.io/deploymentMode",
            cfg.deploy.default_deployment_mode,
        )
        all_ready = True
        if mode == "Serverless":
            ksvc = self.server.try_get(
                "serving.knative.dev/v1/Service", namespace,
                f"{name}-predictor",
            )
            ready = bool(
                ksvc
                and get_condition(ksvc.get("status", {}), "Ready")
                and get_condition(ksvc["status"], "Ready")["status"] == "True"
            )
            set_condition(
                status, "PredictorReady", "True" if ready else "False",
                reason="" if ready else "KsvcNotReady",
            )
            all_ready &= ready
        else:
            for comp in ("predictor", "transformer", "explainer"):
                dep_name = f"{name}-{comp}"
                if (gvk_of({"apiVersion": "apps/v1", "kind": "Deployment"}),
                        dep_name) not in applied_keys:
                    continue
                ready, reason = self._deployment_ready(namespace, dep_name)
                set_condition(
                    status,
                    comp.capitalize() + "Ready",
                    "True" if ready else "False",
                    reason=reason,
                )
                all_ready &= ready

        # OCI ImageVolume advisory (reference controller.go:774-831):
        # oci:// models served through the modelcar sidecar on clusters
        # where the native ImageVolume feature is available get a
        # non-blocking advisory condition nudging migration.
        if uri and uri.startswith(("oci://", "oci+native://")):
            iv_available = getattr(cfg.deploy, "image_volume_available",
                                   False)
            set_condition(
                status, "OCIImageVolumeAdvisory",
                "True" if iv_available else "False",
                reason="ImageVolumeAvailable" if iv_available
                else "ModelcarFallback",
                message=(
                    "cluster supports native OCI ImageVolumes; consider "
                    "migrating off the modelcar sidecar"
                    if iv_available else
                    "oci:// model served via modelcar sidecar"
                ),
            )

        set_condition(status, "IngressReady", "True")
        set_condition(
            status, "Ready", "True" if all_ready else "False",
            reason="" if all_ready else "ComponentsNotReady",
        )
        if all_ready:
            dom = cfg.ingress
            from kserve_amd.controlplane.configmap import render_domain

            status["url"] = (
                f"{dom.url_scheme}://{render_domain(dom, name, namespace)}"
            )
            status["address"] = {
                "url": f"http://{name}-predictor.{namespace}.svc.cluster.local"
            }
        self._update_status(obj, status)
        if not all_ready:
            # PropagateModelStatus-style requeue until pods are ready
            return Result(requeue_after=0.05)
        return None

    def _prune(self, namespace: str, owner_name: str, keep: set) -> None:
        """Remove previously-created objects no longer desired (e.g. the
        canary Deployment after promotion, the HPA after maxReplicas is
        cleared)."""
        for g in MANAGED_GVKS:
            for o in self.server.list(
                g,
                namespace,
                label_selector={
                    "serving.kserve.io/inferenceservice": owner_name
                },
            ):
                k = (g, o["metadata"]["name"])
                if k not in keep:
                    delete_if_exists(
                        self.server, g, namespace, o["metadata"]["name"]
                    )

    # -- wiring ------------------------------------------------------------
    def build(self) -> Controller:
        c = Controller(
            self.server,
            ISVC_GVK,
            self.reconcile,
            owned_gvks=("apps/v1/Deployment", "serving.knative.dev/v1/Service"),
            owner_label="serving.kserve.io/inferenceservice",
        )
        c.start_watches()
        return c


class FakeDeploymentController:
    """Envtest stand-in for the kubelet/deployment controller: marks every
    Deployment available (status.availableReplicas = spec.replicas). Lets a
    ControllerManager converge an ISVC to Ready in tests, the role the
    reference's envtest suites play by patching status by hand."""

    def __init__(self, server, delay_updates: int = 0):
        self.server = server
        # number of reconciles to leave a deployment unavailable first
        # (exercises the requeue-until-ready path)
        self.delay = delay_updates
        self._seen: Dict[Tuple[str, str], int] = {}

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        ns, name = key
        dep = self.server.try_get("apps/v1/Deployment", ns, name)
        if dep is None:
            return None
        n = self._seen.get(key, 0)
        self._seen[key] = n + 1
        if n < self.delay:
            return Result(requeue_after=0.01)
        want = dep.get("spec", {}).get("replicas", 1)
        if dep.get("status", {}).get("availableReplicas") != want:
            dep["status"] = {
                "availableReplicas": want,
                "readyReplicas": want,
                "replicas": want,
            }
            self.server.update_status(dep)
        return None

    def build(self) -> Controller:
        c = Controller(self.server, "apps/v1/Deployment", self.reconcile)
        c.start_watches()
        return c
