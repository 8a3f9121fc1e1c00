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

        try:
            arr = np.asarray(instances, dtype=float)
        except (TypeError, ValueError):
            # ragged payloads (e.g. the ART convention [image, label])
            arr = instances
        return await self.explainer.explain(arr, predict_fn)


class SquareAttackExplainer(Explainer):
    """Native black-box square attack (the algorithm behind the reference's
    artexplainer: random square perturbations, accepted when they push the
    prediction away from the true label — artserver/model.py:74-96). No ART
    dependency; works against any predictor returning class labels or
    per-class scores."""

    def __init__(
        self,
        nb_classes: int,
        max_iter: int = 100,
        eps: float = 0.3,
        seed: int = 0,
    ):
        self.nb_classes = int(nb_classes)
        self.max_iter = int(max_iter)
        self.eps = float(eps)
        self.seed = seed

    @staticmethod
    def _to_scores(pred, nb_classes) -> np.ndarray:
        arr = np.asarray(pred, dtype=float)
        if arr.ndim == 0 or (arr.ndim == 1 and arr.size == 1):
            one_hot = np.zeros(nb_classes)
            one_hot[int(arr)] = 1.0
            return one_hot
        return arr.reshape(-1)

    async def explain(self, instances: np.ndarray, predict_fn) -> Dict:
        # payload convention (reference artserver): [image, label]
        x = np.asarray(instances[0], dtype=float)
        label = int(np.asarray(instances[1]).reshape(-1)[0])
        rng = np.random.default_rng(self.seed)
        flat = x.reshape(-1)
        n = flat.size
        side = int(np.sqrt(n)) if int(np.sqrt(n)) ** 2 == n else None

        async def score(v: np.ndarray) -> np.ndarray:
            p = await predict_fn([v.reshape(x.shape).tolist()])
            return self._to_scores(p[0], self.nb_classes)

        orig_scores = await score(flat)
        orig_pred = int(np.argmax(orig_scores))
        best = flat.copy()
        best_margin = orig_scores[label] - np.max(
            np.delete(orig_scores, label)
        )
        # square-attack loop: progressively smaller random squares of +-eps
        for it in range(self.max_iter):
            frac = max(0.05, 0.5 * (1 - it / self.max_iter))
            cand = best.copy()
            if side is not None:
                h = max(1, int(side * frac))
                r0 = rng.integers(0, side - h + 1)
                c0 = rng.integers(0, side - h + 1)
                sq = (slice(r0, r0 + h), slice(c0, c0 + h))
                img = cand.reshape(side, side)
                img[sq] = img[sq] + rng.choice([-self.eps, self.eps])
            else:
                w = max(1, int(n * frac))
                i0 = rng.integers(0, n - w + 1)
                cand[i0 : i0 + w] += rng.choice([-self.eps, self.eps])
            s = await score(cand)
            margin = s[label] - np.max(np.delete(s, label))
            if margin < best_margin:
                best, best_margin = cand, margin
                if margin < 0:
                    break
        adv_scores = await score(best)
        adv = best.reshape(x.shape)
        return {
            "explanations": {
                "adversarial_example": [adv.tolist()],
                "L2 error": float(np.linalg.norm((adv - x).reshape(-1))),
                "adversarial_prediction": int(np.argmax(adv_scores)),
                "prediction": orig_pred,
            }
        }
