"""Model registry (reference parity: python/kserve/kserve/model_repository.py:23-80
plus the Triton-style repository extension, protocol/model_repository_extension.py:23)."""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Union

from kserve_amd.errors import ModelNotFound
from kserve_amd.logging import logger
from kserve_amd.model import BaseModel

MODEL_MOUNT_DIRS = "/mnt/models"


class ModelRepository:
    def __init__(self, models_dir: str = MODEL_MOUNT_DIRS):
        self.models: Dict[str, BaseModel] = {}
        self.models_dir = models_dir

    def load_models(self) -> None:
        """Scan models_dir for per-model subdirectories (multi-model serving)."""
        if not os.path.exists(self.models_dir):
            return
        for name in os.listdir(self.models_dir):
            d = os.path.join(self.models_dir, name)
            if os.path.isdir(d):
                self.load_model(name)

    def set_models_dir(self, models_dir: str):
        self.models_dir = models_dir

    def get_model(self, name: str) -> Optional[BaseModel]:
        return self.models.get(name)

    def get_models(self) -> Dict[str, BaseModel]:
        return self.models

    def is_model_ready(self, name: str) -> bool:
        model = self.get_model(name)
        return bool(model and model.ready)

    def update(self, model: BaseModel):
        self.models[model.name] = model

    def update_handle(self, model: BaseModel, name: Optional[str] = None):
        self.models[name or model.name] = model

    def load(self, name: str) -> bool:
        """Dynamic load hook for the repository extension; override for
        framework-specific loading."""
        return self.load_model(name)

    def load_model(self, name: str) -> bool:
        model = self.get_model(name)
        if model is None:
            return False
        if not model.ready:
            model.load()
        return model.ready

    def unload(self, name: str):
        model = self.models.pop(name, None)
        if model is None:
            raise ModelNotFound(name)
        model.stop()
        logger.info("Model %s unloaded", name)

    # -- Triton-style repository index (V2 /v2/repository/index) -----------
    def index(self) -> List[dict]:
        out = []
        for name, model in self.models.items():
            out.append(
                {
                    "name": name,
                    "state": "READY" if model.ready else "UNAVAILABLE",
                    "reason": "",
                }
            )
        return out
