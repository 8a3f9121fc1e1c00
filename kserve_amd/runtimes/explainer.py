"""Explainer support: the :explain verb with a pluggable explainer.

Reference parity: explainer component (v1beta1 explainer.go, artexplainer
sample — ART SquareAttack). The ART library is not installed in this
offline image; ExplainerModel provides the serving contract (forward
predict to the predictor, run an Explainer over it) and a built-in
occlusion-sensitivity explainer that needs only numpy.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from kserve_amd.errors import InvalidInput
from kserve_amd.model import Model, PredictorConfig


class Explainer:
    async def explain(self, instances: np.ndarray, predict_fn) -> Dict:
        raise NotImplementedError


class OcclusionExplainer(Explainer):
    """Feature-importance via occlusion: zero one feature at a time and
    measure the prediction change (dependency-free stand-in for ART)."""

    def __init__(self, baseline: float = 0.0):
        self.baseline = baseline

    async def explain(self, instances: np.ndarray, predict_fn) -> Dict:
        base_pred = np.asarray(await predict_fn(instances.tolist()))
        importances = []
        for row_idx, row in enumerate(instances):
            row_imp = []
            for f in range(len(row)):
                perturbed = row.copy()
                perturbed[f] = self.baseline
                pred = np.asarray(await predict_fn([perturbed.tolist()]))[0]
                delta = np.abs(
                    np.asarray(base_pred[row_idx], dtype=float) - np.asarray(pred, dtype=float)
                )
                row_imp.append(float(np.max(delta)))
            importances.append(row_imp)
        return {"explanations": {"importances": importances}}


class ARTExplainer(Explainer):
    """Adversarial-Robustness-Toolbox SquareAttack (reference
    artexplainer/artserver) — import-gated: ART is not in this image."""

    def __init__(self, **kwargs):
        try:
            import art  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "adversarial-robustness-toolbox is not installed in this "
                "offline image; use OcclusionExplainer or install ART"
            ) from e


class ExplainerModel(Model):
    """Explainer component: ``:predict`` forwards to the predictor,
    ``:explain`` runs the explainer against it (reference explainer_art.go
    arg injection --predictor_host)."""

    def __init__(
        self,
        name: str,
        predictor_config: PredictorConfig,
        explainer: Optional[Explainer] = None,
    ):
        super().__init__(name, predictor_config=predictor_config)
        self.explainer = explainer or OcclusionExplainer()
        self.ready = True

    async def explain(self, payload, headers=None):
        instances = payload.get("instances")
        if instances is None:
            raise InvalidInput('Expected "instances"')

        async def predict_fn(batch):
            result = await self._forward_predict({"instances": batch}, headers)
            return result["predictions"]

        return await self.explainer.explain(np.asarray(instances, dtype=float), predict_fn)
