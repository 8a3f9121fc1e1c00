"""LLMInferenceService (v1alpha2-style) types + workload rendering.

Reference parity: pkg/apis/serving/v1alpha2/llm_inference_service_types.go —
LLMInferenceServiceSpec (:85-127), WorkloadSpec (:131-192),
KVCacheOffloadingSpec (:213-290), SchedulerSpec/EPP (:471-533),
ParallelismSpec (:733-759), LLMModelSpec+LoRA (:307-372) — and the workload
renderers (llmisvc/workload_single_node.go:43-199, workload_multi_node.go
LWS, config_merge.go arg generation). Ours renders the NATIVE engine's CLI
(kserve_amd.runtimes.huggingfaceserver) instead of `vllm serve`.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class ParallelismSpec:
    """reference :733-759."""

    tensor: int = 1
    pipeline: int = 1
    data: int = 1
    data_local: Optional[int] = None
    data_rpc_port: int = 5555
    expert: bool = False

    @property
    def ranks_per_replica(self) -> int:
        return self.tensor * self.pipeline

    @property
    def is_multi_node(self) -> bool:
        return self.pipeline > 1 or (
            self.data_local is not None and self.data_local < self.data
        )


@dataclass
class KVCacheOffloadingSpec:
    """reference :213-290 (cpu tier + cascading fs tiers)."""

    cpu_bytes_to_use: int = 0
    eviction_policy: str = "lru"  # lru | arc
    filesystem_tiers: List[Dict] = field(default_factory=list)  # emptyDir/pvc


@dataclass
class LoRASpec:
    adapters: List[Dict] = field(default_factory=list)  # {name, storageUri}


@dataclass
class LLMModelSpec:
    uri: str = ""
    name: Optional[str] = None
    lora: Optional[LoRASpec] = None


@dataclass
class WorkloadSpec:
    replicas: int = 1
    parallelism: ParallelismSpec = field(default_factory=ParallelismSpec)
    kv_cache_offloading: Optional[KVCacheOffloadingSpec] = None
    max_model_len: int = 8192
    max_num_seqs: int = 256
    resources: Dict = field(default_factory=dict)


@dataclass
class SchedulerSpec:
    """EPP endpoint-picker deployment (reference :471-533)."""

    enabled: bool = True
    grpc_port: int = 9002
    health_port: int = 9003


@dataclass
class TracingSpec:
    """reference :707-731 -> OTel env injection (llmisvc/tracing.go)."""

    enabled: bool = False
    otlp_endpoint: str = ""
    sample_rate: float = 0.05


@dataclass
class LLMInferenceServiceSpec:
    model: LLMModelSpec
    workload: WorkloadSpec = field(default_factory=WorkloadSpec)
    prefill: Optional[WorkloadSpec] = None  # disaggregated prefill pool
    scheduler: Optional[SchedulerSpec] = None
    tracing: Optional[TracingSpec] = None


@dataclass
class LLMInferenceService:
    name: str
    namespace: str = "default"
    spec: LLMInferenceServiceSpec = None


def render_engine_args(spec: LLMInferenceServiceSpec, workload: WorkloadSpec) -> List[str]:
    """The native-engine CLI the reference renders for vLLM
    (config-llm-worker-data-parallel.yaml:188-199 flags)."""
    p = workload.parallelism
    args = [
        "--model_dir=/mnt/models",
        f"--model_name={spec.model.name or 'model'}",
        "--backend=engine",
        f"--tensor-parallel-size={p.tensor}",
        f"--max_model_len={workload.max_model_len}",
        f"--max_num_seqs={workload.max_num_seqs}",
    ]
    if p.data > 1:
        args += [
            f"--data-parallel-size={p.data}",
            f"--data-parallel-rpc-port={p.data_rpc_port}",
        ]
        if p.data_local is not None:
            args.append(f"--data-parallel-size-local={p.data_local}")
    if p.expert:
        args.append("--enable-expert-parallel")
    return args


def render_workload(
    llm: LLMInferenceService,
    role: str = "decode",
    image: str = "kserve-amd/huggingfaceserver:latest",
) -> Dict:
    """Single-node Deployment or multi-node LeaderWorkerSet manifest."""
    spec = llm.spec
    workload = spec.prefill if role == "prefill" else spec.workload
    if workload is None:
        raise ValueError(f"no {role} workload")
    p = workload.parallelism
    env = [
        {"name": "HSA_ENABLE_IPC_MODE_LEGACY", "value": "0"},
        {"name": "MASTER_ADDR", "value": "127.0.0.1"},
    ]
    if spec.tracing and spec.tracing.enabled:
        env += [
            {"name": "OTEL_EXPORTER_OTLP_ENDPOINT", "value": spec.tracing.otlp_endpoint},
            {"name": "OTEL_TRACES_SAMPLER", "value": "parentbased_traceidratio"},
            {"name": "OTEL_TRACES_SAMPLER_ARG", "value": str(spec.tracing.sample_rate)},
        ]
    container = {
        "name": "kserve-container",
        "image": image,
        "command": ["python", "-m", "kserve_amd.runtimes.huggingfaceserver"],
        "args": render_engine_args(spec, workload),
        "env": env,
        "resources": workload.resources
        or {"limits": {"amd.com/gpu": str(p.ranks_per_replica)}},
    }
    volumes = []
    if workload.kv_cache_offloading:
        kv = workload.kv_cache_offloading
        container["args"].append(
            f"--kv-offload-bytes={kv.cpu_bytes_to_use}"
        )
        for i, tier in enumerate(kv.filesystem_tiers):
            volumes.append({"name": f"kv-tier-{i}", **tier})
    name = f"{llm.name}-{role}"
    if not p.is_multi_node:
        return {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "metadata": {"name": name, "namespace": llm.namespace},
            "spec": {
                "replicas": workload.replicas,
                "selector": {"matchLabels": {"app": name}},
                "template": {
                    "metadata": {"labels": {"app": name, "llm-role": role}},
                    "spec": {"containers": [container], "volumes": volumes},
                },
            },
        }
    # multi-node: LeaderWorkerSet (reference workload_multi_node.go:42-139)
    # with the torchrun-rendezvous health probes (reference wires Ray
    # health_check.py probes into the multinode runtime; ours probes the
    # torch.distributed TCPStore instead — kserve_amd/parallel/health.py)
    group_size = p.pipeline
    probe_cmd = ["python", "-m", "kserve_amd.parallel.health"]
    liveness = {
        "exec": {"command": probe_cmd + ["gpu"]},
        "periodSeconds": 30,
        "failureThreshold": 3,
    }
    readiness = {
        "exec": {"command": probe_cmd + ["store"]},
        "periodSeconds": 10,
    }
    leader_container = {**container, "livenessProbe": liveness,
                        "readinessProbe": readiness}
    worker_container = {
        **container,
        "livenessProbe": liveness,
        "readinessProbe": readiness,
        # workers wait for the leader's rendezvous endpoint to listen
        "startupProbe": {
            "exec": {"command": probe_cmd + ["master"]},
            "periodSeconds": 5,
            "failureThreshold": 60,
        },
    }
    return {
        "apiVersion": "leaderworkerset.x-k8s.io/v1",
        "kind": "LeaderWorkerSet",
        "metadata": {"name": name, "namespace": llm.namespace},
        "spec": {
            "replicas": workload.replicas,
            "leaderWorkerTemplate": {
                "size": group_size,
                "leaderTemplate": {
                    "metadata": {"labels": {"app": name, "role": "leader"}},
                    "spec": {"containers": [leader_container],
                             "volumes": volumes},
                },
                "workerTemplate": {
                    "metadata": {"labels": {"app": name, "role": "worker"}},
                    "spec": {"containers": [worker_container],
                             "volumes": volumes},
                },
            },
        },
    }


def render_scheduler(llm: LLMInferenceService, image="kserve-amd/endpoint-picker:latest") -> Optional[Dict]:
    """EPP deployment (reference scheduler.go:74-388)."""
    if llm.spec.scheduler is None or not llm.spec.scheduler.enabled:
        return None
    s = llm.spec.scheduler
    return {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {
            "name": f"{llm.name}-epp",
            "namespace": llm.namespace,
        },
        "spec": {
            "replicas": 1,
            "selector": {"matchLabels": {"app": f"{llm.name}-epp"}},
            "template": {
                "metadata": {"labels": {"app": f"{llm.name}-epp"}},
                "spec": {
                    "containers": [
                        {
                            "name": "main",
                            "image": image,
                            "command": [
                                "python", "-m",
                                "kserve_amd.agent.endpoint_picker",
                            ],
                            "args": [
                                "--endpoints",
                                f"http://{llm.name}-decode.{llm.namespace}:80",
                                "--port", str(s.grpc_port),
                                "--http-port", str(s.health_port),
                            ],
                            "ports": [
                                {"containerPort": s.grpc_port, "name": "grpc"},
                                {"containerPort": s.health_port, "name": "health"},
                            ],
                        }
                    ]
                },
            },
        },
    }


def reconcile_llm(llm: LLMInferenceService) -> Dict[str, object]:
    """decode (+ optional prefill) workloads + scheduler, mirroring the
    reference's reconcile pipeline (llmisvc/controller.go:258-298)."""
    out: Dict[str, object] = {"decode": render_workload(llm, "decode")}
    if llm.spec.prefill is not None:
        out["prefill"] = render_workload(llm, "prefill")
    sched = render_scheduler(llm)
    if sched is not None:
        out["scheduler"] = sched
    return out


# -- scaling: WVA / KEDA (reference v1alpha2 ScalingSpec :548-690) -----------

def render_scaling(llm: LLMInferenceService, scaling: Dict) -> List[Dict]:
    """ScalingSpec → manifests. ``wva`` renders a
    WorkloadVariantAutoscaler CR (llm-d WVA: per-variant profiles with
    SLO targets); ``keda`` renders a ScaledObject on the decode
    Deployment; ``hpa`` (default) is handled by the caller's HPA path.
    Reference: ScalingSpec/WVASpec/KEDAScalingSpec
    (llm_inference_service_types.go:548-690), KEDA Fallback floor
    (:668-672)."""
    out: List[Dict] = []
    name = f"{llm.name}-decode"
    wva = scaling.get("wva")
    if wva:
        out.append(
            {
                "apiVersion": "llmd.ai/v1alpha1",
                "kind": "WorkloadVariantAutoscaler",
                "metadata": {"name": name, "namespace": llm.namespace},
                "spec": {
                    "scaleTargetRef": {
                        "apiVersion": "apps/v1",
                        "kind": "Deployment",
                        "name": name,
                    },
                    "minReplicas": wva.get("minReplicas", 1),
                    "maxReplicas": wva.get("maxReplicas", 8),
                    "sloTargets": {
                        "ttftMs": wva.get("ttftMs", 500),
                        "tpotMs": wva.get("tpotMs", 50),
                    },
                    "profiles": wva.get("profiles", []),
                },
            }
        )
    keda = scaling.get("keda")
    if keda:
        so = {
            "apiVersion": "keda.sh/v1alpha1",
            "kind": "ScaledObject",
            "metadata": {"name": name, "namespace": llm.namespace},
            "spec": {
                "scaleTargetRef": {
                    "apiVersion": "apps/v1",
                    "kind": "Deployment",
                    "name": name,
                },
                "minReplicaCount": keda.get("minReplicas", 1),
                "maxReplicaCount": keda.get("maxReplicas", 8),
                "triggers": keda.get("triggers", []),
            },
        }
        if keda.get("fallback"):
            # replica floor during metric outages (:668-672)
            so["spec"]["fallback"] = {
                "failureThreshold": keda["fallback"].get("failureThreshold", 3),
                "replicas": keda["fallback"].get("replicas", 1),
            }
        out.append(so)
    return out


def render_inference_pool(llm: LLMInferenceService) -> Dict:
    """Gateway API Inference Extension InferencePool selecting the decode
    pods, with the EPP as the endpoint-picker extension (reference
    scheduler.go:221-387 InferencePool v1/v1alpha2 wiring)."""
    s = llm.spec.scheduler or SchedulerSpec()
    return {
        "apiVersion": "inference.networking.x-k8s.io/v1alpha2",
        "kind": "InferencePool",
        "metadata": {"name": llm.name, "namespace": llm.namespace},
        "spec": {
            "targetPortNumber": 8080,
            "selector": {"app": f"{llm.name}-decode"},
            "extensionRef": {
                "name": f"{llm.name}-epp",
                "portNumber": s.grpc_port,
                "failureMode": "FailClose",
            },
        },
    }
