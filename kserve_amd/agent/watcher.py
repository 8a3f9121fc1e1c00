"""Multi-model agent: model-config watcher + puller.

Reference parity: pkg/agent/watcher.go:65-196 (fsnotify watch of
/mnt/configs modelconfig -> diff -> per-model serialized ops),
puller.go:61-160 (download channel), downloader.go:41, syncer.go:36
(startup sync). Python version polls mtime (no fsnotify dependency) and
serializes per-model operations through asyncio.
"""

from __future__ import annotations

import asyncio
import json
import os
import shutil
from dataclasses import dataclass
from typing import Awaitable, Callable, Dict, List, Optional

from kserve_amd.logging import logger
from kserve_amd.storage import Storage

MODEL_CONFIG_FILE = "modelconfig.json"


@dataclass
class ModelSpec:
    storage_uri: str
    framework: str = ""
    memory: str = ""

    @classmethod
    def from_dict(cls, d: Dict) -> "ModelSpec":
        spec = d.get("modelSpec") or d.get("spec") or {}
        return cls(
            storage_uri=spec.get("storageUri", ""),
            framework=spec.get("framework", ""),
            memory=spec.get("memory", ""),
        )


class ModelConfigWatcher:
    """Watches a modelconfig file and converges local models to it.

    on_load(name, local_dir, spec) / on_unload(name) hooks plug into a
    ModelRepository (load/unload) — the reference's puller+server contract.
    """

    def __init__(
        self,
        config_dir: str,
        model_dir: str,
        on_load: Callable[[str, str, ModelSpec], Awaitable[None]],
        on_unload: Callable[[str], Awaitable[None]],
        poll_interval_s: float = 1.0,
    ):
        self.config_path = os.path.join(config_dir, MODEL_CONFIG_FILE)
        self.model_dir = model_dir
        self.on_load = on_load
        self.on_unload = on_unload
        self.poll_interval_s = poll_interval_s
        self.current: Dict[str, ModelSpec] = {}
        self._mtime: Optional[float] = None
        self._task: Optional[asyncio.Task] = None
        self._stopping = False

    # -- one reconciliation pass -------------------------------------------
    @staticmethod
    def parse_config(raw: str) -> Dict[str, ModelSpec]:
        entries = json.loads(raw) if raw.strip() else []
        out = {}
        for e in entries:
            name = e.get("modelName") or e.get("name")
            if name:
                out[name] = ModelSpec.from_dict(e)
        return out

    async def sync_once(self) -> bool:
        """Returns True if a change was processed."""
        try:
            mtime = os.path.getmtime(self.config_path)
        except FileNotFoundError:
            return False
        if self._mtime is not None and mtime == self._mtime:
            return False
        self._mtime = mtime
        with open(self.config_path) as f:
            desired = self.parse_config(f.read())
        await self.apply(desired)
        return True

    async def apply(self, desired: Dict[str, ModelSpec]):
        # unload removed / changed models first (reference ConfigsDelta.Process)
        for name in list(self.current):
            if (
                name not in desired
                or desired[name].storage_uri != self.current[name].storage_uri
            ):
                await self._unload(name)
        for name, spec in desired.items():
            if name not in self.current:
                await self._load(name, spec)

    async def _load(self, name: str, spec: ModelSpec):
        local = os.path.join(self.model_dir, name)
        os.makedirs(local, exist_ok=True)
        try:
            await asyncio.get_running_loop().run_in_executor(
                None, Storage.download, spec.storage_uri, local
            )
            await self.on_load(name, local, spec)
            self.current[name] = spec
            logger.info("Agent loaded model %s from %s", name, spec.storage_uri)
        except Exception:
            logger.exception("Agent failed to load model %s", name)

    async def _unload(self, name: str):
        try:
            await self.on_unload(name)
        except Exception:
            logger.exception("Agent failed to unload model %s", name)
        self.current.pop(name, None)
        local = os.path.join(self.model_dir, name)
        if os.path.isdir(local):
            shutil.rmtree(local, ignore_errors=True)
        logger.info("Agent unloaded model %s", name)

    # -- background loop ----------------------------------------------------
    async def start(self):
        self._stopping = False
        self._task = asyncio.create_task(self._loop())

    async def _loop(self):
        while not self._stopping:
            try:
                await self.sync_once()
            except Exception:
                logger.exception("Agent watcher sync error")
            await asyncio.sleep(self.poll_interval_s)

    async def stop(self):
        self._stopping = True
        if self._task:
            self._task.cancel()
