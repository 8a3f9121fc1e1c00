"""LocalModelCache: node-level warm model cache.

Reference parity: pkg/controller/v1alpha1/{localmodel,localmodelnode} —
LocalModelCache CR (v1alpha1 localmodelcache :36-81) declares models to
pre-download per node group; the node agent (localmodelnode/controller.go:527,
launchJob :117) keeps the node's cache directory converged (download via the
storage-initializer machinery, folder checks, reconcile loop). Ours runs the
same convergence in-process against a hostPath-style cache dir.
"""

from __future__ import annotations

import asyncio
import os
import shutil
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from kserve_amd.logging import logger
from kserve_amd.storage import Storage


@dataclass
class LocalModelCacheSpec:
    """CR: which model, which node groups, how big."""

    name: str
    source_model_uri: str
    model_size: str = ""
    node_groups: List[str] = field(default_factory=lambda: ["default"])


@dataclass
class LocalModelStatus:
    name: str
    state: str  # Pending | Downloading | Ready | Failed
    message: str = ""


class LocalModelNodeAgent:
    """Per-node agent converging the cache dir to the declared cache CRs."""

    def __init__(
        self,
        cache_dir: str,
        node_group: str = "default",
        reconcile_interval_s: float = 60.0,
    ):
        self.cache_dir = cache_dir
        self.node_group = node_group
        self.reconcile_interval_s = reconcile_interval_s
        self.desired: Dict[str, LocalModelCacheSpec] = {}
        self.status: Dict[str, LocalModelStatus] = {}
        os.makedirs(cache_dir, exist_ok=True)

    # -- CR management -------------------------------------------------------
    def apply(self, spec: LocalModelCacheSpec):
        if self.node_group in spec.node_groups:
            self.desired[spec.name] = spec

    def delete(self, name: str):
        self.desired.pop(name, None)

    def model_path(self, name: str) -> str:
        return os.path.join(self.cache_dir, "models", name)

    def is_cached(self, name: str) -> bool:
        p = self.model_path(name)
        return os.path.isdir(p) and bool(os.listdir(p))

    # -- reconcile -----------------------------------------------------------
    async def reconcile_once(self) -> Dict[str, LocalModelStatus]:
        # download missing models
        for name, spec in list(self.desired.items()):
            if self.is_cached(name):
                self.status[name] = LocalModelStatus(name, "Ready")
                continue
            self.status[name] = LocalModelStatus(name, "Downloading")
            dest = self.model_path(name)
            os.makedirs(dest, exist_ok=True)
            try:
                await asyncio.get_running_loop().run_in_executor(
                    None, Storage.download, spec.source_model_uri, dest
                )
                self.status[name] = LocalModelStatus(name, "Ready")
                logger.info("LocalModel %s cached at %s", name, dest)
            except Exception as e:
                self.status[name] = LocalModelStatus(name, "Failed", str(e))
                logger.exception("LocalModel %s download failed", name)
        # remove models no longer declared (reference folder GC)
        models_root = os.path.join(self.cache_dir, "models")
        if os.path.isdir(models_root):
            for name in os.listdir(models_root):
                if name not in self.desired:
                    shutil.rmtree(os.path.join(models_root, name), ignore_errors=True)
                    self.status.pop(name, None)
                    logger.info("LocalModel %s evicted", name)
        return dict(self.status)

    async def run(self):
        while True:
            try:
                await self.reconcile_once()
            except Exception:
                logger.exception("LocalModel reconcile error")
            await asyncio.sleep(self.reconcile_interval_s)


def mount_for_isvc(agent: LocalModelNodeAgent, storage_uri: str) -> Optional[str]:
    """If an ISVC's storage uri matches a cached model, serve from the local
    cache path instead of re-downloading (reference: ISVC pods mount the
    local PV)."""
    for name, spec in agent.desired.items():
        if spec.source_model_uri == storage_uri and agent.is_cached(name):
            return agent.model_path(name)
    return None
