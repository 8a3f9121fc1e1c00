"""Unified predictive model server: one runtime serving sklearn, xgboost
and lightgbm artifacts, with a multi-model repository mode.

Reference parity: python/predictiveserver/predictiveserver/model.py:26-110
(framework dispatch) and model_repository.py (directory-scan multi-model).
Framework may be given explicitly or inferred from the artifact extension.
"""

from __future__ import annotations

import os
from typing import Optional

from kserve_amd.model import Model
from kserve_amd.model_repository import ModelRepository

FRAMEWORK_SKLEARN = "sklearn"
FRAMEWORK_XGBOOST = "xgboost"
FRAMEWORK_LIGHTGBM = "lightgbm"
SUPPORTED_FRAMEWORKS = (FRAMEWORK_SKLEARN, FRAMEWORK_XGBOOST, FRAMEWORK_LIGHTGBM)

_EXT_TO_FRAMEWORK = {
    ".joblib": FRAMEWORK_SKLEARN,
    ".pkl": FRAMEWORK_SKLEARN,
    ".pickle": FRAMEWORK_SKLEARN,
    ".ubj": FRAMEWORK_XGBOOST,
    ".bst": FRAMEWORK_XGBOOST,  # .bst is ambiguous; xgboost wins like ref
    ".json": FRAMEWORK_XGBOOST,
    ".txt": FRAMEWORK_LIGHTGBM,
}


def detect_framework(model_dir: str) -> Optional[str]:
    for entry in sorted(os.listdir(model_dir)):
        fw = _EXT_TO_FRAMEWORK.get(os.path.splitext(entry)[1].lower())
        if fw:
            return fw
    return None


class PredictiveServerModel(Model):
    """Dispatch to the framework-specific runtime (sklearn/xgb/lgb)."""

    def __init__(
        self,
        name: str,
        model_dir: str,
        framework: Optional[str] = None,
        nthread: int = 1,
    ):
        super().__init__(name)
        self.model_dir = model_dir
        framework = (framework or detect_framework(model_dir) or "").lower()
        if framework not in SUPPORTED_FRAMEWORKS:
            raise ValueError(
                f"Unsupported framework: {framework!r}. "
                f"Supported: {', '.join(SUPPORTED_FRAMEWORKS)}"
            )
        self.framework = framework
        self.nthread = nthread
        self._model = self._create_framework_model()

    def _create_framework_model(self) -> Model:
        if self.framework == FRAMEWORK_SKLEARN:
            from kserve_amd.runtimes.sklearnserver import SKLearnModel

            return SKLearnModel(self.name, self.model_dir)
        if self.framework == FRAMEWORK_XGBOOST:
            from kserve_amd.runtimes.xgbserver import XGBoostModel

            return XGBoostModel(self.name, self.model_dir, self.nthread)
        from kserve_amd.runtimes.lgbserver import LightGBMModel

        return LightGBMModel(self.name, self.model_dir, self.nthread)

    def load(self) -> bool:
        self.ready = self._model.load()
        return self.ready

    def predict(self, payload, headers=None):
        return self._model.predict(payload, headers)


class PredictiveServerModelRepository(ModelRepository):
    """Multi-model serving: each subdirectory of ``models_dir`` is loaded
    as its own PredictiveServerModel (framework auto-detected)."""

    def __init__(self, models_dir: str, framework: Optional[str] = None,
                 nthread: int = 1):
        super().__init__()
        self.models_dir = models_dir
        self.framework = framework
        self.nthread = nthread
        self.load_models()

    def load_models(self):
        for name in sorted(os.listdir(self.models_dir)):
            d = os.path.join(self.models_dir, name)
            if not os.path.isdir(d):
                continue
            try:
                model = PredictiveServerModel(
                    name, d, framework=self.framework, nthread=self.nthread
                )
                model.load()
                self.update(model)
            except Exception:  # framework lib absent / bad artifact
                from kserve_amd.logging import logger

                logger.exception("predictiveserver: failed to load %s", name)


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    parser = build_arg_parser()
    parser.add_argument("--framework", default=None,
                        choices=list(SUPPORTED_FRAMEWORKS))
    parser.add_argument("--nthread", type=int, default=1)
    parser.add_argument("--models_dir", default=None,
                        help="multi-model mode: serve every subdirectory")
    args = parser.parse_args(argv)
    if args.models_dir:
        repo = PredictiveServerModelRepository(
            args.models_dir, framework=args.framework, nthread=args.nthread
        )
        ModelServer(http_port=args.http_port, registered_models=repo).start([])
    else:
        model = PredictiveServerModel(
            args.model_name, args.model_dir, framework=args.framework,
            nthread=args.nthread,
        )
        model.load()
        ModelServer(http_port=args.http_port).start([model])


if __name__ == "__main__":
    main()
