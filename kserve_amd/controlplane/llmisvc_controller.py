"""Live LLMInferenceService controller.

Reference parity: pkg/controller/v1alpha2/llmisvc/controller.go:258-298 —
the reconcile pipeline: presets merged via baseRefs (config_merge.go) →
router validation (router*.go) → decode workload (single-node Deployment or
multi-node LeaderWorkerSet) + optional prefill pool → scheduler (EPP
deployment + service) → TLS secret → status conditions. This module runs
that pipeline against the APIServer interface with the same apply/prune/
status machinery as the ISVC controller.

CR shape (camelCase spec mirroring v1alpha2 llm_inference_service_types.go):

    apiVersion: serving.kserve.io/v1alpha2
    kind: LLMInferenceService
    spec:
      baseRefs: [kserve-config-llm-template, ...]
      model: {name: meta/llama-3-8b, uri: hf://...}
      workload: {replicas, parallelism: {tensor, pipeline, data, expert}, ...}
      prefill: {...}            # optional disaggregated prefill pool
      router: {route: ..., gateway: ..., scheduler: ...}
      tls: {selfSigned: true}
"""

from __future__ import annotations

import copy
from typing import Dict, List, Optional, Tuple

from kserve_amd.controlplane.controller import (
    Controller,
    Result,
    create_or_update,
    delete_if_exists,
    set_condition,
)
from kserve_amd.controlplane.llmisvc import (
    KVCacheOffloadingSpec,
    LLMInferenceService,
    LLMInferenceServiceSpec,
    LLMModelSpec,
    ParallelismSpec,
    SchedulerSpec,
    TracingSpec,
    WorkloadSpec,
    render_inference_pool,
    render_scaling,
    render_scheduler,
    render_workload,
)
from kserve_amd.controlplane.llmisvc_config import (
    ConfigMergeError,
    render_config,
    render_tls_secret,
    validate_router,
)

LLM_GVK = "serving.kserve.io/v1alpha2/LLMInferenceService"
LLMCFG_GVK = "serving.kserve.io/v1alpha2/LLMInferenceServiceConfig"

MANAGED = (
    "apps/v1/Deployment",
    "leaderworkerset.x-k8s.io/v1/LeaderWorkerSet",
    "v1/Service",
    "v1/Secret",
    "keda.sh/v1alpha1/ScaledObject",
    "llmd.ai/v1alpha1/WorkloadVariantAutoscaler",
    "inference.networking.x-k8s.io/v1alpha2/InferencePool",
)


def _workload_from_dict(src: Dict) -> WorkloadSpec:
    par = src.get("parallelism", {}) or {}
    kv = None
    if src.get("kvCacheOffloading"):
        k = src["kvCacheOffloading"]
        kv = KVCacheOffloadingSpec(
            cpu_bytes_to_use=k.get("cpuBytesToUse", 0),
            eviction=k.get("eviction", "lru"),
            filesystem_tiers=k.get("filesystemTiers", []) or [],
        )
    return WorkloadSpec(
        replicas=src.get("replicas", 1),
        parallelism=ParallelismSpec(
            tensor=par.get("tensor", 1),
            pipeline=par.get("pipeline", 1),
            data=par.get("data", 1),
            data_local=par.get("dataLocal"),
            data_rpc_port=par.get("dataRpcPort", 5555),
            expert=par.get("expert", False),
        ),
        kv_cache_offloading=kv,
        max_model_len=src.get("maxModelLen", 8192),
        max_num_seqs=src.get("maxNumSeqs", 256),
        resources=src.get("resources", {}) or {},
    )


def llm_from_manifest(obj: Dict, merged_spec: Dict) -> LLMInferenceService:
    md = obj["metadata"]
    model = merged_spec.get("model", {}) or {}
    sched = None
    if merged_spec.get("scheduler") is not None:
        ssrc = merged_spec["scheduler"] or {}
        sched = SchedulerSpec(
            enabled=ssrc.get("enabled", True),
            grpc_port=ssrc.get("grpcPort", 9002),
            health_port=ssrc.get("healthPort", 9003),
        )
    tracing = None
    if merged_spec.get("tracing"):
        t = merged_spec["tracing"]
        tracing = TracingSpec(
            enabled=bool(t.get("enabled")),
            otlp_endpoint=t.get("otlpEndpoint", ""),
            sample_rate=float(t.get("sampleRate", 0.05)),
        )
    return LLMInferenceService(
        name=md["name"],
        namespace=md.get("namespace", "default"),
        spec=LLMInferenceServiceSpec(
            model=LLMModelSpec(
                name=model.get("name", ""),
                uri=model.get("uri", ""),
            ),
            workload=_workload_from_dict(merged_spec.get("workload", {}) or {}),
            prefill=(
                _workload_from_dict(merged_spec["prefill"])
                if merged_spec.get("prefill")
                else None
            ),
            scheduler=sched,
            tracing=tracing,
        ),
    )


class LLMInferenceServiceController:
    def __init__(self, server):
        self.server = server

    def _extra_configs(self, namespace: str) -> Dict[str, Dict]:
        """Cluster LLMInferenceServiceConfig CRs extend the built-in preset
        catalog (config_merge.go resolves baseRefs against CRs first)."""
        out: Dict[str, Dict] = {}
        for o in self.server.list(LLMCFG_GVK, namespace) + self.server.list(
            LLMCFG_GVK, ""
        ):
            out[o["metadata"]["name"]] = o.get("spec", {}) or {}
        return out

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        namespace, name = key
        obj = self.server.try_get(LLM_GVK, namespace, name)
        if obj is None:
            return None
        raw_spec = copy.deepcopy(obj.get("spec", {}) or {})
        base_refs = raw_spec.pop("baseRefs", []) or []
        status = copy.deepcopy(obj.get("status", {}) or {})

        # presets + templating + router validation
        try:
            merged = render_config(
                name, namespace, raw_spec, base_refs,
                extra_configs=self._extra_configs(namespace),
            )
        except ConfigMergeError as e:
            set_condition(status, "Ready", "False", reason="ConfigMergeError",
                          message=str(e))
            self._update_status(obj, status)
            return None
        violations = validate_router(merged.get("router"))
        if violations:
            set_condition(
                status, "RouterValid", "False", reason="InvalidRouter",
                message="; ".join(violations),
            )
            set_condition(status, "Ready", "False", reason="InvalidRouter")
            self._update_status(obj, status)
            return None
        set_condition(status, "RouterValid", "True")

        llm = llm_from_manifest(obj, merged)
        applied = set()

        def apply(m: Dict):
            m["metadata"].setdefault("namespace", namespace)
            m["metadata"].setdefault("labels", {})[
                "serving.kserve.io/llminferenceservice"
            ] = name
            a = create_or_update(self.server, m, owner=obj)
            applied.add((f"{a['apiVersion']}/{a['kind']}",
                         a["metadata"]["name"]))

        workloads = {"decode": render_workload(llm, "decode")}
        if llm.spec.prefill is not None:
            workloads["prefill"] = render_workload(llm, "prefill")
        for role, m in workloads.items():
            apply(m)
            apply(
                {
                    "apiVersion": "v1",
                    "kind": "Service",
                    "metadata": {"name": f"{name}-{role}",
                                 "namespace": namespace},
                    "spec": {
                        "selector": {"app": f"{name}-{role}"},
                        "ports": [{"name": "http", "port": 80,
                                   "targetPort": 8080}],
                    },
                }
            )
        sched = render_scheduler(llm)
        if sched is not None:
            apply(sched)
            apply(render_inference_pool(llm))
            apply(
                {
                    "apiVersion": "v1",
                    "kind": "Service",
                    "metadata": {"name": f"{name}-epp",
                                 "namespace": namespace},
                    "spec": {
                        "selector": {"app": f"{name}-epp"},
                        "ports": [
                            {"name": "grpc",
                             "port": llm.spec.scheduler.grpc_port},
                        ],
                    },
                }
            )
        for m in render_scaling(llm, merged.get("scaling", {}) or {}):
            apply(m)
        if (raw_spec.get("tls") or {}).get("selfSigned"):
            # one-time self-signed pair; regenerate only if absent
            if self.server.try_get("v1/Secret", namespace, f"{name}-tls") is None:
                apply(render_tls_secret(f"{name}-tls", namespace,
                                        f"{name}-decode"))
            else:
                applied.add(("v1/Secret", f"{name}-tls"))

        # prune
        for g in MANAGED:
            for o in self.server.list(
                g, namespace,
                label_selector={"serving.kserve.io/llminferenceservice": name},
            ):
                k = (g, o["metadata"]["name"])
                if k not in applied:
                    delete_if_exists(self.server, g, namespace,
                                     o["metadata"]["name"])

        # status: decode workload availability drives Ready
        ready = True
        for role in workloads:
            wname = f"{name}-{role}"
            dep = self.server.try_get("apps/v1/Deployment", namespace, wname)
            lws = self.server.try_get(
                "leaderworkerset.x-k8s.io/v1/LeaderWorkerSet", namespace, wname
            )
            if dep is not None:
                want = dep.get("spec", {}).get("replicas", 1)
                have = dep.get("status", {}).get("availableReplicas", 0)
                ok = have >= max(1, want)
            elif lws is not None:
                ok = (
                    lws.get("status", {}).get("readyReplicas", 0)
                    >= lws.get("spec", {}).get("replicas", 1)
                )
            else:
                ok = False
            set_condition(
                status, f"{role.capitalize()}Ready",
                "True" if ok else "False",
                reason="" if ok else "WorkloadNotReady",
            )
            ready &= ok
        set_condition(status, "Ready", "True" if ready else "False",
                      reason="" if ready else "WorkloadsNotReady")
        if ready:
            status["url"] = (
                f"http://{name}-decode.{namespace}.svc.cluster.local"
            )
        self._update_status(obj, status)
        if not ready:
            return Result(requeue_after=0.05)
        return None

    def _update_status(self, obj: Dict, status: Dict) -> None:
        if obj.get("status", {}) == status:
            return
        newobj = copy.deepcopy(obj)
        newobj["status"] = status
        self.server.update_status(newobj)

    def build(self) -> Controller:
        c = Controller(
            self.server,
            LLM_GVK,
            self.reconcile,
            owned_gvks=("apps/v1/Deployment",
                        "leaderworkerset.x-k8s.io/v1/LeaderWorkerSet"),
            owner_label="serving.kserve.io/llminferenceservice",
        )
        c.start_watches()
        return c
