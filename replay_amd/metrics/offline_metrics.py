"""Batch metric runner + Experiment table.

Parity with reference replay/metrics/offline_metrics.py:12 (``OfflineMetrics``
computing many metrics over one recommendation frame) and
replay/metrics/experiment.py:7 (``Experiment`` results table with
``add_result``:158 and ``compare``:178).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Union

import pandas as pd

from .base_metric import Metric, MetricsDataFrameLike, MetricsReturnType
from .metrics import CategoricalDiversity, Coverage, Novelty, Surprisal, Unexpectedness, _TrainAwareMetric


class OfflineMetrics:
    """Compute a list of metrics in one call.

    >>> from replay_amd.metrics import NDCG, Recall
    >>> import pandas as pd
    >>> recs = pd.DataFrame({"query_id": [1, 1], "item_id": [10, 11], "rating": [0.9, 0.8]})
    >>> gt = pd.DataFrame({"query_id": [1], "item_id": [11]})
    >>> out = OfflineMetrics([NDCG([2]), Recall([2])])(recs, gt)
    >>> round(out["Recall@2"], 3)
    1.0
    """

    def __init__(
        self,
        metrics: List[Metric],
        query_column: str = "query_id",
        item_column: str = "item_id",
        rating_column: str = "rating",
        category_column: str = "category_id",
        allow_caching: bool = True,
    ) -> None:
        self.metrics = metrics
        self.query_column = query_column
        self.item_column = item_column
        self.rating_column = rating_column
        # OfflineMetrics owns the column naming (reference offline_metrics.py)
        for metric in metrics:
            metric.query_column = query_column
            metric.rating_column = rating_column
            if not isinstance(metric, CategoricalDiversity):
                metric.item_column = item_column

    def __call__(
        self,
        recommendations: MetricsDataFrameLike,
        ground_truth: MetricsDataFrameLike,
        train: Optional[MetricsDataFrameLike] = None,
        base_recommendations: Optional[Union[MetricsDataFrameLike, Dict[str, MetricsDataFrameLike]]] = None,
    ) -> MetricsReturnType:
        result: MetricsReturnType = {}
        for metric in self.metrics:
            if isinstance(metric, Unexpectedness):
                if base_recommendations is None:
                    raise ValueError("Unexpectedness requires base_recommendations")
                if isinstance(base_recommendations, dict) and not any(
                    isinstance(v, (int, float, str)) for v in list(base_recommendations)[:0]
                ) and all(isinstance(v, pd.DataFrame) for v in base_recommendations.values()):
                    for name, base in base_recommendations.items():
                        out = metric(recommendations, base)
                        # reference key layout: Metric_MODEL@k
                        result.update({
                            f"{key.split('@')[0]}_{name}@{key.split('@')[1]}": value
                            for key, value in out.items()
                        })
                else:
                    result.update(metric(recommendations, base_recommendations))
            elif isinstance(metric, _TrainAwareMetric):
                result.update(metric(recommendations, train=train))
            elif isinstance(metric, CategoricalDiversity):
                result.update(metric(recommendations))
            else:
                result.update(metric(recommendations, ground_truth))
        return result


class Experiment:
    """Results table comparing models over a fixed eval setup
    (reference experiment.py:7)."""

    def __init__(
        self,
        metrics: List[Metric],
        ground_truth: MetricsDataFrameLike,
        train: Optional[MetricsDataFrameLike] = None,
        base_recommendations: Optional[MetricsDataFrameLike] = None,
        query_column: str = "query_id",
        item_column: str = "item_id",
        rating_column: str = "rating",
    ) -> None:
        self.ground_truth = ground_truth
        self.train = train
        self.base_recommendations = base_recommendations
        self.metrics = OfflineMetrics(
            metrics, query_column=query_column, item_column=item_column, rating_column=rating_column
        )
        self.results = pd.DataFrame()

    def add_result(self, name: str, recommendations: MetricsDataFrameLike) -> None:
        out = self.metrics(
            recommendations,
            self.ground_truth,
            train=self.train,
            base_recommendations=self.base_recommendations,
        )
        row = pd.DataFrame(out, index=[name])
        self.results = pd.concat([self.results[~self.results.index.isin([name])], row])

    def compare(self, name: str) -> pd.DataFrame:
        """Relative difference of every model vs the named one."""
        if name not in self.results.index:
            raise ValueError(f"No results for {name}")
        baseline = self.results.loc[name]
        others = self.results.drop(index=name)
        return (others - baseline) / baseline
