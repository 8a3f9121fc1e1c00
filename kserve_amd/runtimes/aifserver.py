"""AI-Fairness explainer: bias metrics over a predictor's outputs.

Reference parity: python/aiffairness/aifserver/model.py:59-91 — the
:explain verb takes instances + predictions and returns
BinaryLabelDatasetMetric values. The metrics are computed natively
(numpy) instead of via aif360 (not in this image); definitions follow
the aif360 documentation:
- base_rate            P(label = favorable)
- statistical_parity_difference  P(fav | unprivileged) - P(fav | privileged)
- disparate_impact     P(fav | unprivileged) / P(fav | privileged)
- consistency          1 - mean_i |y_i - mean_{j in kNN(i)} y_j| (k=5)
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from kserve_amd.errors import InvalidInput
from kserve_amd.model import Model, PredictorConfig


def _group_mask(
    features: np.ndarray, feature_names: List[str], groups: List[Dict]
) -> np.ndarray:
    """Rows matching ANY of the group dicts ({feature_name: value})."""
    mask = np.zeros(features.shape[0], dtype=bool)
    for g in groups:
        m = np.ones(features.shape[0], dtype=bool)
        for name, value in g.items():
            idx = feature_names.index(name)
            m &= features[:, idx] == value
        mask |= m
    return mask


def _consistency(features: np.ndarray, labels: np.ndarray, k: int = 5) -> float:
    n = features.shape[0]
    k = min(k, n - 1)
    if k <= 0:
        return 1.0
    # pairwise L2 distances (datasets here are explain-payload sized)
    d = np.linalg.norm(features[:, None, :] - features[None, :, :], axis=-1)
    np.fill_diagonal(d, np.inf)
    knn = np.argsort(d, axis=1)[:, :k]
    return float(1.0 - np.mean(np.abs(labels - labels[knn].mean(axis=1))))


class AIFFairnessModel(Model):
    """Fairness explainer component (:predict forwards; :explain scores)."""

    def __init__(
        self,
        name: str,
        feature_names: List[str],
        label_names: List[str],
        favorable_label: float,
        unfavorable_label: float,
        privileged_groups: List[Dict],
        unprivileged_groups: List[Dict],
        predictor_config: Optional[PredictorConfig] = None,
    ):
        super().__init__(name, predictor_config=predictor_config)
        self.feature_names = list(feature_names)
        self.label_names = list(label_names)
        self.favorable_label = favorable_label
        self.unfavorable_label = unfavorable_label
        self.privileged_groups = privileged_groups
        self.unprivileged_groups = unprivileged_groups
        self.ready = True

    async def explain(self, payload: Dict, headers=None) -> Dict:
        instances = payload.get("instances")
        if instances is None:
            raise InvalidInput('Expected "instances"')
        features = np.asarray(instances, dtype=float)
        predictions = payload.get("outputs")
        if predictions is None:
            # no precomputed outputs: ask the predictor
            result = await self._forward_predict(
                {"instances": features.tolist()}, headers
            )
            predictions = result["predictions"]
        labels = np.asarray(predictions, dtype=float).reshape(-1)
        fav = labels == self.favorable_label

        priv = _group_mask(features, self.feature_names, self.privileged_groups)
        unpriv = _group_mask(
            features, self.feature_names, self.unprivileged_groups
        )
        p_priv = float(fav[priv].mean()) if priv.any() else float("nan")
        p_unpriv = float(fav[unpriv].mean()) if unpriv.any() else float("nan")
        disparate = p_unpriv / p_priv if p_priv else float("inf")
        return {
            "predictions": np.asarray(predictions).tolist(),
            "metrics": {
                "base_rate": float(fav.mean()),
                "consistency": [_consistency(features, labels)],
                "disparate_impact": disparate,
                "num_instances": float(labels.size),
                "num_negatives": float((labels == self.unfavorable_label).sum()),
                "num_positives": float(fav.sum()),
                "statistical_parity_difference": p_unpriv - p_priv,
            },
        }
