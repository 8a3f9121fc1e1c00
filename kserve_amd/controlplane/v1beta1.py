"""InferenceService v1beta1 API types: spec, defaulting, validation.

Reference parity: pkg/apis/serving/v1beta1 — InferenceServiceSpec
(inference_service.go:24-43), PredictorSpec one-of over frameworks
(predictor.go:32-87), WorkerSpec (:105-117), ComponentExtensionSpec
(component.go:82-132), LoggerSpec (:94-125), Batcher (:145-155), defaulting
(inference_service_defaults.go) and validation
(inference_service_validation.go) as pure functions.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional

SUPPORTED_STORAGE_SCHEMES = (
    "gs", "s3", "pvc", "file", "https", "http", "hdfs", "webhdfs", "hf",
    "oci", "oci+native",
)

FRAMEWORKS = (
    "sklearn", "xgboost", "lightgbm", "huggingface", "pmml", "paddle",
    "onnx", "tensorflow", "pytorch", "triton", "model",
)


class ValidationError(ValueError):
    pass


@dataclass
class ModelFormat:
    name: str
    version: Optional[str] = None


@dataclass
class PredictorModelSpec:
    """The `model:` form (runtime auto-selection; predictor_model.go:88-223)."""

    model_format: ModelFormat = None
    storage_uri: Optional[str] = None
    runtime: Optional[str] = None
    protocol_version: str = "v1"
    resources: Dict[str, Dict[str, str]] = field(default_factory=dict)
    args: List[str] = field(default_factory=list)
    env: Dict[str, str] = field(default_factory=dict)
    image: Optional[str] = None


@dataclass
class FrameworkSpec:
    """Per-framework shorthand (predictor_sklearn.go etc.)."""

    storage_uri: Optional[str] = None
    runtime_version: Optional[str] = None
    protocol_version: str = "v1"
    resources: Dict[str, Dict[str, str]] = field(default_factory=dict)


@dataclass
class WorkerSpec:
    """Multi-node (predictor.go:105-117)."""

    size: int = 1
    pipeline_parallel_size: Optional[int] = None
    tensor_parallel_size: Optional[int] = None


@dataclass
class LoggerSpec:
    mode: str = "all"  # all | request | response
    url: Optional[str] = None


@dataclass
class BatcherSpec:
    max_batch_size: int = 32
    max_latency_ms: int = 5000


@dataclass
class ComponentExtensionSpec:
    min_replicas: int = 1
    max_replicas: int = 0  # 0 = unlimited
    scale_target: Optional[int] = None
    scale_metric: Optional[str] = None  # cpu | memory | concurrency | rps
    canary_traffic_percent: Optional[int] = None
    timeout_seconds: Optional[int] = None
    logger: Optional[LoggerSpec] = None
    batcher: Optional[BatcherSpec] = None
    service_account_name: Optional[str] = None
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)


@dataclass
class PredictorSpec(ComponentExtensionSpec):
    model: Optional[PredictorModelSpec] = None
    sklearn: Optional[FrameworkSpec] = None
    xgboost: Optional[FrameworkSpec] = None
    lightgbm: Optional[FrameworkSpec] = None
    huggingface: Optional[FrameworkSpec] = None
    pmml: Optional[FrameworkSpec] = None
    paddle: Optional[FrameworkSpec] = None
    triton: Optional[FrameworkSpec] = None
    tensorflow: Optional[FrameworkSpec] = None
    pytorch: Optional[FrameworkSpec] = None
    onnx: Optional[FrameworkSpec] = None
    containers: List[Dict] = field(default_factory=list)  # custom predictor
    worker: Optional[WorkerSpec] = None

    def implementations(self) -> List[str]:
        out = []
        for fw in ("sklearn", "xgboost", "lightgbm", "huggingface", "pmml",
                   "paddle", "triton", "tensorflow", "pytorch", "onnx"):
            if getattr(self, fw) is not None:
                out.append(fw)
        # `model` is its own implementation only when no framework shorthand
        # is present (the defaulter promotes shorthands into model form)
        if self.model is not None and not out:
            out.append("model")
        if self.containers:
            out.append("custom")
        return out

    @property
    def framework(self) -> Optional[str]:
        impls = self.implementations()
        if not impls:
            return None
        if impls[0] == "model":
            return self.model.model_format.name
        return impls[0]

    @property
    def storage_uri(self) -> Optional[str]:
        impls = self.implementations()
        if not impls:
            return None
        if impls[0] == "model":
            return self.model.storage_uri
        if impls[0] == "custom":
            return None
        return getattr(self, impls[0]).storage_uri


@dataclass
class TransformerSpec(ComponentExtensionSpec):
    containers: List[Dict] = field(default_factory=list)


@dataclass
class ExplainerSpec(ComponentExtensionSpec):
    art: Optional[FrameworkSpec] = None
    containers: List[Dict] = field(default_factory=list)


@dataclass
class InferenceServiceSpec:
    predictor: PredictorSpec
    transformer: Optional[TransformerSpec] = None
    explainer: Optional[ExplainerSpec] = None


@dataclass
class InferenceService:
    name: str
    namespace: str = "default"
    spec: InferenceServiceSpec = None
    annotations: Dict[str, str] = field(default_factory=dict)
    labels: Dict[str, str] = field(default_factory=dict)

    @property
    def deployment_mode(self) -> str:
        # reference: isvcutils.GetDeploymentMode (annotation > default)
        return self.annotations.get(
            "serving.kserve.io/deploymentMode", "RawDeployment"
        )


_NAME_RE = re.compile(r"^[a-z]([-a-z0-9]*[a-z0-9])?$")


def default_inference_service(isvc: InferenceService) -> InferenceService:
    """Mutating defaulter (reference InferenceServiceDefaulter)."""
    p = isvc.spec.predictor
    if p.min_replicas is None or p.min_replicas < 0:
        p.min_replicas = 1
    impls = p.implementations()
    # promote framework shorthand to model form with format name
    if impls and impls[0] != "model" and impls[0] != "custom" and p.model is None:
        fw = impls[0]
        spec = getattr(p, fw)
        p.model = PredictorModelSpec(
            model_format=ModelFormat(name=fw),
            storage_uri=spec.storage_uri,
            protocol_version=spec.protocol_version or "v1",
            resources=spec.resources,
        )
    return isvc


def validate_inference_service(isvc: InferenceService) -> None:
    """Raises ValidationError (reference InferenceServiceValidator)."""
    if not _NAME_RE.match(isvc.name or ""):
        raise ValidationError(
            f"invalid InferenceService name {isvc.name!r} (RFC1035 label)"
        )
    p = isvc.spec.predictor
    impls = p.implementations()
    if len(impls) == 0:
        raise ValidationError("predictor must specify an implementation")
    if len(impls) > 1:
        raise ValidationError(
            f"predictor must specify exactly one implementation, got {impls}"
        )
    uri = p.storage_uri
    if uri and "://" in uri:
        scheme = uri.split("://", 1)[0]
        if scheme not in SUPPORTED_STORAGE_SCHEMES:
            raise ValidationError(f"unsupported storage scheme {scheme!r}")
    ext = p
    if ext.canary_traffic_percent is not None and not (
        0 <= ext.canary_traffic_percent <= 100
    ):
        raise ValidationError("canaryTrafficPercent must be in [0, 100]")
    if p.min_replicas < 0:
        raise ValidationError("minReplicas must be >= 0")
    if p.max_replicas and p.max_replicas < p.min_replicas:
        raise ValidationError("maxReplicas must be >= minReplicas")
    # odd GPU counts rejected (reference validation)
    for comp in (p,):
        res = (p.model.resources if p.model else {}) or {}
        gpus = (res.get("limits") or {}).get("nvidia.com/gpu") or (
            res.get("limits") or {}
        ).get("amd.com/gpu")
        if gpus is not None:
            g = int(gpus)
            if g > 1 and g % 2 == 1:
                raise ValidationError("multi-GPU count must be even")
