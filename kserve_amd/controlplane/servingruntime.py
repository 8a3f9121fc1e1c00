"""ServingRuntime catalog + runtime auto-selection.

Reference parity: pkg/apis/serving/v1alpha1/servingruntime_types.go:151-202
(ServingRuntimeSpec) and the selection algorithm
ModelSpec.GetSupportingRuntimes / RuntimeSupportsModel
(predictor_model.go:88-223): namespace runtimes before cluster runtimes,
filtered by model format + protocol + autoSelect, stable-sorted by
per-format priority.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class SupportedModelFormat:
    name: str
    version: Optional[str] = None
    auto_select: bool = False
    priority: int = 1


@dataclass
class ServingRuntime:
    name: str
    supported_model_formats: List[SupportedModelFormat]
    container: Dict  # pod container template for kserve-container
    protocol_versions: List[str] = field(default_factory=lambda: ["v1", "v2"])
    disabled: bool = False
    multi_model: bool = False
    cluster_scoped: bool = True
    workers: bool = False  # supports multi-node WorkerSpec

    def supports(self, model_format: str, protocol: str, multinode: bool = False) -> bool:
        if self.disabled:
            return False
        if protocol and protocol not in self.protocol_versions:
            return False
        if multinode and not self.workers:
            return False
        return any(
            f.name == model_format and f.auto_select
            for f in self.supported_model_formats
        )

    def priority_for(self, model_format: str) -> int:
        for f in self.supported_model_formats:
            if f.name == model_format:
                return f.priority
        return 0


def select_runtime(
    model_format: str,
    protocol: str,
    namespace_runtimes: List[ServingRuntime],
    cluster_runtimes: List[ServingRuntime],
    explicit_runtime: Optional[str] = None,
    multinode: bool = False,
) -> ServingRuntime:
    all_runtimes = list(namespace_runtimes) + list(cluster_runtimes)
    if explicit_runtime:
        for rt in all_runtimes:
            if rt.name == explicit_runtime:
                if rt.disabled:
                    raise LookupError(f"runtime {explicit_runtime} is disabled")
                return rt
        raise LookupError(f"runtime {explicit_runtime} not found")
    ns = sorted(
        [r for r in namespace_runtimes if r.supports(model_format, protocol, multinode)],
        key=lambda r: -r.priority_for(model_format),
    )
    cl = sorted(
        [r for r in cluster_runtimes if r.supports(model_format, protocol, multinode)],
        key=lambda r: -r.priority_for(model_format),
    )
    candidates = ns + cl
    if not candidates:
        raise LookupError(
            f"no ServingRuntime supports modelFormat={model_format} "
            f"protocol={protocol}"
        )
    return candidates[0]


def default_cluster_runtimes(image_prefix: str = "kserve-amd") -> List[ServingRuntime]:
    """The config/runtimes catalog equivalent (reference §2.4): our native
    runtime images for each model format."""

    def rt(name, formats, module, protocols=("v1", "v2"), workers=False):
        return ServingRuntime(
            name=name,
            supported_model_formats=[
                SupportedModelFormat(name=f, auto_select=True) for f in formats
            ],
            protocol_versions=list(protocols),
            workers=workers,
            container={
                "name": "kserve-container",
                "image": f"{image_prefix}/{name}:latest",
                "command": ["python", "-m", module],
                "args": [
                    "--model_name={{.Name}}",
                    "--model_dir=/mnt/models",
                    "--http_port=8080",
                    "--grpc_port=8081",
                ],
            },
        )

    return [
        rt("kserve-amd-sklearnserver", ["sklearn"], "kserve_amd.runtimes.sklearnserver"),
        rt("kserve-amd-xgbserver", ["xgboost"], "kserve_amd.runtimes.xgbserver"),
        rt("kserve-amd-lgbserver", ["lightgbm"], "kserve_amd.runtimes.lgbserver"),
        rt("kserve-amd-paddleserver", ["paddle"], "kserve_amd.runtimes.paddleserver"),
        rt("kserve-amd-pmmlserver", ["pmml"], "kserve_amd.runtimes.pmmlserver"),
        ServingRuntime(
            name="kserve-amd-predictiveserver",
            supported_model_formats=[
                SupportedModelFormat(name=f, auto_select=True, priority=0)
                for f in ("sklearn", "xgboost", "lightgbm", "onnx")
            ],
            protocol_versions=["v1", "v2"],
            container={
                "name": "kserve-container",
                "image": f"{image_prefix}/kserve-amd-predictiveserver:latest",
                "command": ["python", "-m",
                            "kserve_amd.runtimes.predictiveserver"],
                "args": [
                    "--model_name={{.Name}}",
                    "--model_dir=/mnt/models",
                    "--http_port=8080",
                    "--grpc_port=8081",
                ],
            },
        ),
        rt("kserve-amd-autogluonserver", ["autogluon"], "kserve_amd.runtimes.autogluonserver"),
        rt(
            "kserve-amd-huggingfaceserver",
            ["huggingface"],
            "kserve_amd.runtimes.huggingfaceserver",
            protocols=("v1", "v2", "openai"),
            workers=True,
        ),
        # the LLM-dedicated runtime (vllmserver row of the reference
        # catalog): same module, engine backend forced
        rt(
            "kserve-amd-llmserver",
            ["huggingface"],
            "kserve_amd.runtimes.huggingfaceserver",
            protocols=("openai", "v2"),
            workers=True,
        ),
    ]
