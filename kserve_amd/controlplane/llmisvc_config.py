"""LLMInferenceService preset system: baseRefs config-merge + template
substitution + router validation + self-signed TLS + graceful shutdown.

Reference parity: pkg/controller/v1alpha2/llmisvc/config_merge.go —
``baseRefs`` name LLMInferenceServiceConfig preset CRs whose specs are
merged in order under the service's own spec (:689-722), with Go-template
variable substitution over the merged manifest (:723-765: ``{{ .Name }}``,
``{{ .Namespace }}``, ``{{ .Spec.Model.Name }}``, ``{{ .GlobalConfig.* }}``,
``{{ ChildName .Name "-suffix" }}``); the vLLM ``--shutdown-timeout`` is
computed from the pod's terminationGracePeriodSeconds (:766-788). Router
validation per router*.go (managed route XOR ref'd routes/gateways); TLS
self-signed workload certs per workload_tls_self_signed.go.

Preset catalog mirrors config/llmisvcconfig/*.yaml (11 files) with our
native-engine image/args in place of the llm-d CUDA artifacts.
"""

from __future__ import annotations

import base64
import copy
import os
import re
import subprocess
import tempfile
from typing import Dict, List, Optional


class ConfigMergeError(ValueError):
    pass


# ---------------------------------------------------------------------------
# Preset catalog (config/llmisvcconfig equivalents, MI355X-native images)
# ---------------------------------------------------------------------------

PRESETS: Dict[str, Dict] = {
    # base template: single-node decode workload running the native engine
    "kserve-config-llm-template": {
        "workload": {
            "replicas": 1,
            "maxModelLen": 8192,
            "maxNumSeqs": 256,
            "image": "kserve-amd/huggingfaceserver:latest",
            "env": [{"name": "HSA_ENABLE_IPC_MODE_LEGACY", "value": "0"}],
        },
    },
    "kserve-config-llm-decode-worker-data-parallel": {
        "workload": {
            "parallelism": {"data": 8, "dataLocal": 8, "dataRpcPort": 5555},
            "args": [
                "--data-parallel-size={{ .Spec.Workload.Parallelism.Data }}",
                "--data-parallel-rpc-port={{ .Spec.Workload.Parallelism.DataRpcPort }}",
            ],
        },
    },
    "kserve-config-llm-prefill-template": {
        "prefill": {
            "replicas": 1,
            "image": "kserve-amd/huggingfaceserver:latest",
            "args": ["--role=prefill"],
        },
    },
    "kserve-config-llm-scheduler": {
        "scheduler": {
            "enabled": True,
            "image": "kserve-amd/endpoint-picker:latest",
            "grpcPort": 9002,
            "healthPort": 9003,
        },
    },
    "kserve-config-llm-router-route": {
        "router": {"route": {"http": {"spec": {}}}, "gateway": {}},
    },
    "kserve-config-llm-tokenizer": {
        "tokenizer": {
            "image": "kserve-amd/tokenizer:latest",
            "endpoint": "/tokenize",
        },
    },
    "kserve-config-llm-tracing": {
        "tracing": {
            "enabled": True,
            "otlpEndpoint": "http://{{ .GlobalConfig.OtelCollector }}:4317",
            "sampleRate": "0.05",
        },
    },
}


def _deep_merge(base: Dict, overlay: Dict) -> Dict:
    """Overlay wins; dicts merge recursively; lists REPLACE (the reference
    uses strategic-merge semantics where CR lists replace preset lists)."""
    out = copy.deepcopy(base)
    for k, v in overlay.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def merge_base_refs(
    spec: Dict,
    base_refs: List[str],
    presets: Optional[Dict[str, Dict]] = None,
    extra_configs: Optional[Dict[str, Dict]] = None,
) -> Dict:
    """config_merge.go:689-722: presets merged IN ORDER (later baseRefs win
    over earlier), the service's own spec merged last (wins over all)."""
    catalog = dict(presets or PRESETS)
    if extra_configs:
        catalog.update(extra_configs)
    merged: Dict = {}
    for ref in base_refs:
        preset = catalog.get(ref)
        if preset is None:
            raise ConfigMergeError(f"unknown baseRef {ref!r}")
        merged = _deep_merge(merged, preset)
    return _deep_merge(merged, spec)


# ---------------------------------------------------------------------------
# Template variable substitution (config_merge.go:723-765)
# ---------------------------------------------------------------------------

_VAR_RE = re.compile(r"\{\{\s*([^}]+?)\s*\}\}")


def _lookup(path: str, ctx: Dict):
    """Resolve `.Spec.Model.Name`-style paths case-insensitively against
    nested dicts (camelCase keys)."""
    cur = ctx
    for part in path.lstrip(".").split("."):
        if not isinstance(cur, dict):
            raise ConfigMergeError(f"template path {path!r}: {part!r} not found")
        match = None
        for k in cur:
            if k.lower() == part.lower():
                match = k
                break
        if match is None:
            raise ConfigMergeError(f"template path {path!r}: {part!r} not found")
        cur = cur[match]
    return cur


def _render_expr(expr: str, ctx: Dict) -> str:
    expr = expr.strip()
    # ChildName "a" "-suffix" | ChildName .Name "-suffix"
    if expr.startswith("ChildName"):
        parts = re.findall(r'"([^"]*)"|(\.[\w.]+)', expr[len("ChildName"):])
        vals = []
        for lit, ref in parts:
            vals.append(lit if lit else str(_lookup(ref, ctx)))
        name = "".join(vals)
        # the reference truncates to 63 chars with a hash suffix
        if len(name) > 63:
            import hashlib

            h = hashlib.sha256(name.encode()).hexdigest()[:8]
            name = name[:54] + "-" + h
        return name
    return str(_lookup(expr, ctx))


def substitute_variables(obj, ctx: Dict):
    """Walk the merged spec and substitute {{ ... }} in every string."""
    if isinstance(obj, str):
        return _VAR_RE.sub(lambda m: _render_expr(m.group(1), ctx), obj)
    if isinstance(obj, dict):
        return {k: substitute_variables(v, ctx) for k, v in obj.items()}
    if isinstance(obj, list):
        return [substitute_variables(v, ctx) for v in obj]
    return obj


def render_config(
    name: str,
    namespace: str,
    spec: Dict,
    base_refs: Optional[List[str]] = None,
    global_config: Optional[Dict] = None,
    presets: Optional[Dict[str, Dict]] = None,
    extra_configs: Optional[Dict[str, Dict]] = None,
) -> Dict:
    """Full pipeline: merge baseRefs then substitute template variables."""
    merged = merge_base_refs(
        spec, base_refs or [], presets=presets, extra_configs=extra_configs
    )
    ctx = {
        "Name": name,
        "Namespace": namespace,
        "Spec": merged,
        "GlobalConfig": global_config or {},
    }
    return substitute_variables(merged, ctx)


# ---------------------------------------------------------------------------
# Graceful shutdown (config_merge.go:766-788)
# ---------------------------------------------------------------------------

def shutdown_timeout_seconds(termination_grace_period: int) -> int:
    """The engine's --shutdown-timeout: grace period minus drain headroom
    (the reference reserves 15s for pod teardown, floor of 5s)."""
    return max(5, int(termination_grace_period) - 15)


# ---------------------------------------------------------------------------
# Router validation (router.go / router_group.go semantics)
# ---------------------------------------------------------------------------

def validate_router(router: Optional[Dict]) -> List[str]:
    """Returns a list of violations (empty = valid):
    - route.http.spec (managed route) is mutually exclusive with
      route.http.refs (user-provided routes)
    - gateway.refs requires no managed gateway spec and vice versa
    - scheduler pool refs must not combine with a managed scheduler
    """
    errs: List[str] = []
    if not router:
        return errs
    http = (router.get("route") or {}).get("http") or {}
    if http.get("spec") is not None and http.get("refs"):
        errs.append("router.route.http: spec and refs are mutually exclusive")
    gw = router.get("gateway") or {}
    if gw.get("spec") is not None and gw.get("refs"):
        errs.append("router.gateway: spec and refs are mutually exclusive")
    if http.get("refs") and not gw.get("refs"):
        errs.append(
            "router.route.http.refs requires router.gateway.refs (a route "
            "must attach to a referenced gateway)"
        )
    sched = router.get("scheduler") or {}
    if sched.get("pool") is not None and sched.get("spec") is not None:
        errs.append("router.scheduler: pool ref and managed spec are exclusive")
    return errs


# ---------------------------------------------------------------------------
# Self-signed TLS (workload_tls_self_signed.go)
# ---------------------------------------------------------------------------

def generate_self_signed_cert(
    common_name: str, dns_names: List[str], days: int = 365
) -> Dict[str, bytes]:
    """Self-signed cert/key pair for in-cluster TLS between router and
    workloads. Uses the system openssl (always present in our images)."""
    with tempfile.TemporaryDirectory() as d:
        key = os.path.join(d, "tls.key")
        crt = os.path.join(d, "tls.crt")
        san = ",".join(f"DNS:{n}" for n in dns_names) or f"DNS:{common_name}"
        subprocess.run(
            [
                "openssl", "req", "-x509", "-newkey", "rsa:2048",
                "-keyout", key, "-out", crt, "-days", str(days),
                "-nodes", "-subj", f"/CN={common_name}",
                "-addext", f"subjectAltName={san}",
            ],
            check=True,
            capture_output=True,
        )
        with open(key, "rb") as f:
            key_pem = f.read()
        with open(crt, "rb") as f:
            crt_pem = f.read()
    return {"tls.key": key_pem, "tls.crt": crt_pem}


def render_tls_secret(name: str, namespace: str, service_name: str) -> Dict:
    """Kubernetes TLS Secret manifest with a fresh self-signed pair covering
    the workload service's cluster-local DNS names."""
    dns = [
        service_name,
        f"{service_name}.{namespace}",
        f"{service_name}.{namespace}.svc",
        f"{service_name}.{namespace}.svc.cluster.local",
    ]
    pair = generate_self_signed_cert(dns[0], dns)
    return {
        "apiVersion": "v1",
        "kind": "Secret",
        "type": "kubernetes.io/tls",
        "metadata": {"name": name, "namespace": namespace},
        "data": {
            k: base64.b64encode(v).decode() for k, v in pair.items()
        },
    }
