"""NCIS-weighted precision (reference experimental/metrics/
ncis_precision.py: Precision@K(i) = sum(1_hit * w) / sum(w))."""

import numpy as np

from .base_metric import NCISMetric


class NCISPrecision(NCISMetric):
    def _weighted_user_metric(self, hits: np.ndarray, weights: np.ndarray, k: int) -> float:
        denom = float(weights.sum())
        if denom <= 0:
            return 0.0
        return float((hits * weights).sum() / denom)
