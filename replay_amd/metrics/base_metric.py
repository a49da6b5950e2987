"""Offline metric base machinery.

Parity with reference replay/metrics/base_metric.py:34 (``Metric`` with
``__call__``:111 dispatching dataframe kinds, per-user values via
``_get_metric_value_by_user``:380) and descriptors
(replay/metrics/descriptors.py:35-77).  Pandas/numpy-native.

Call convention (same as reference):
    metric = NDCG([5, 10])
    metric(recommendations, ground_truth)  ->  {"NDCG@5": ..., "NDCG@10": ...}

``recommendations``: DataFrame [query_id, item_id, rating] or dict
{query_id: [item_id, ...]} already ranked.  ``ground_truth``: DataFrame
[query_id, item_id].
"""

from __future__ import annotations

from typing import Dict, List, Mapping, Optional, Sequence, Union

import numpy as np
import pandas as pd


# ---------------------------------------------------------------------------
# aggregation descriptors
# ---------------------------------------------------------------------------
class CalculationDescriptor:
    """How per-user metric values aggregate to the reported number."""

    @property
    def __name__(self) -> str:
        return type(self).__name__

    def cpu(self, values: np.ndarray):  # pragma: no cover
        raise NotImplementedError


class Mean(CalculationDescriptor):
    def cpu(self, values: np.ndarray) -> float:
        return float(np.mean(values)) if len(values) else 0.0


class Median(CalculationDescriptor):
    def cpu(self, values: np.ndarray) -> float:
        return float(np.median(values)) if len(values) else 0.0


class ConfidenceInterval(CalculationDescriptor):
    """Half-width of the normal-approximation confidence interval."""

    def __init__(self, alpha: float = 0.95) -> None:
        self.alpha = alpha

    def cpu(self, values: np.ndarray) -> float:
        from scipy.stats import norm

        if len(values) < 2:
            return 0.0
        quantile = norm.ppf((1 + self.alpha) / 2)
        return float(quantile * values.std(ddof=1) / np.sqrt(len(values)))


class PerUser(CalculationDescriptor):
    def cpu(self, values: np.ndarray):
        return values


# ---------------------------------------------------------------------------
# metric base
# ---------------------------------------------------------------------------
MetricsDataFrameLike = Union[pd.DataFrame, Dict]
MetricsReturnType = Dict[str, Union[float, pd.DataFrame]]


class Metric:
    """Base ranking metric over top-k recommendation lists."""

    def __init__(
        self,
        topk: Union[int, List[int]],
        query_column: str = "query_id",
        item_column: str = "item_id",
        rating_column: str = "rating",
        mode: Optional[CalculationDescriptor] = None,
    ) -> None:
        if isinstance(topk, int):
            topk = [topk]
        for k in topk:
            if not isinstance(k, int) or k <= 0:
                raise ValueError(f"k must be a positive int, got {k}")
        self.topk = sorted(topk)
        self.query_column = query_column
        self.item_column = item_column
        self.rating_column = rating_column
        self._mode = mode if mode is not None else Mean()

    @property
    def __name__(self) -> str:
        return type(self).__name__

    # -- conversion helpers ----------------------------------------------------
    def _recs_to_dict(self, recommendations: MetricsDataFrameLike) -> Dict:
        """query -> ranked list of items (descending rating, stable)."""
        if isinstance(recommendations, dict):
            return recommendations
        df = recommendations
        df = df.sort_values(self.rating_column, ascending=False, kind="stable")
        return df.groupby(self.query_column, sort=False)[self.item_column].apply(list).to_dict()

    def _gt_to_dict(self, ground_truth: MetricsDataFrameLike) -> Dict:
        if isinstance(ground_truth, dict):
            return ground_truth
        return ground_truth.groupby(self.query_column, sort=False)[self.item_column].apply(list).to_dict()

    # -- core ------------------------------------------------------------------
    def __call__(
        self,
        recommendations: MetricsDataFrameLike,
        ground_truth: MetricsDataFrameLike,
        train: Optional[MetricsDataFrameLike] = None,
    ) -> MetricsReturnType:
        recs = self._recs_to_dict(recommendations)
        gt = self._gt_to_dict(ground_truth)
        return self._compute(recs, gt, self._gt_to_dict(train) if train is not None else None)

    def _compute(self, recs: Dict, gt: Dict, train: Optional[Dict]) -> MetricsReturnType:
        queries = list(gt.keys())
        per_k: Dict[int, List[float]] = {k: [] for k in self.topk}
        for q in queries:
            pred = recs.get(q, [])
            truth = gt[q]
            for k in self.topk:
                per_k[k].append(self._get_metric_value_by_user(k, pred, truth))
        result: MetricsReturnType = {}
        for k in self.topk:
            values = np.asarray(per_k[k], dtype=np.float64)
            result.update(self._format_result(k, values, queries))
        return result

    def _format_result(self, k: int, values: np.ndarray, queries) -> "MetricsReturnType":
        """Apply the aggregation mode with the reference's key layout
        ("Metric@k", "Metric-Median@k", "Metric-PerUser@k" -> {query: v})."""
        agg = self._mode.cpu(values)
        if isinstance(self._mode, PerUser):
            return {f"{self.__name__}-PerUser@{k}": dict(zip(queries, agg.tolist()))}
        name = (
            f"{self.__name__}@{k}"
            if isinstance(self._mode, Mean)
            else f"{self.__name__}-{self._mode.__name__}@{k}"
        )
        return {name: agg}

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:  # pragma: no cover
        raise NotImplementedError


class DuplicateWarning(Warning):
    """Duplicate (query, item) pairs in recommendations."""
