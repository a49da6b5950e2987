"""Concrete offline ranking metrics.

Parity with reference replay/metrics/ per-user formulas: HitRate
(hitrate.py:61-73), NDCG (ndcg.py:82-94), MAP (map.py), MRR (mrr.py),
Precision (precision.py), Recall (recall.py), RocAuc (rocauc.py), Coverage
(coverage.py), Novelty (novelty.py), Surprisal (surprisal.py),
Unexpectedness (unexpectedness.py), CategoricalDiversity
(categorical_diversity.py).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Union

import numpy as np
import pandas as pd

from .base_metric import Mean, Metric, MetricsDataFrameLike, MetricsReturnType


class HitRate(Metric):
    """1 if any of the top-k predictions is in ground truth."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        return 1.0 if any(p in truth for p in pred[:k]) else 0.0


class Precision(Metric):
    """Share of top-k predictions that are relevant."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        return sum(1 for p in pred[:k] if p in truth) / k


class Recall(Metric):
    """Share of ground truth captured in top-k."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        return sum(1 for p in pred[:k] if p in truth) / len(truth)


class MRR(Metric):
    """Reciprocal rank of the first relevant prediction in top-k."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        truth = set(ground_truth)
        for rank, p in enumerate(pred[:k]):
            if p in truth:
                return 1.0 / (rank + 1)
        return 0.0


class MAP(Metric):
    """Mean average precision at k (reference map.py formula)."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        hits = 0
        acc = 0.0
        for rank, p in enumerate(pred[:k]):
            if p in truth:
                hits += 1
                acc += hits / (rank + 1)
        return acc / min(k, len(truth))


class NDCG(Metric):
    """Normalized DCG with log2 discount (reference ndcg.py:82-94)."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        dcg = sum(1.0 / np.log2(rank + 2) for rank, p in enumerate(pred[:k]) if p in truth)
        ideal = sum(1.0 / np.log2(rank + 2) for rank in range(min(k, len(truth))))
        return dcg / ideal if ideal > 0 else 0.0


class RocAuc(Metric):
    """AUC over the top-k list: relevant vs non-relevant ordering."""

    @staticmethod
    def _get_metric_value_by_user(k: int, pred: Sequence, ground_truth: Sequence) -> float:
        if not pred or not ground_truth:
            return 0.0
        truth = set(ground_truth)
        labels = [1 if p in truth else 0 for p in pred[:k]]
        n_pos = sum(labels)
        n_neg = len(labels) - n_pos
        if n_pos == 0 or n_neg == 0:
            return 0.0
        # predictions are ranked best-first: count correctly ordered pairs
        auc = 0.0
        seen_neg = 0
        for label in reversed(labels):
            if label == 0:
                seen_neg += 1
            else:
                auc += seen_neg
        return auc / (n_pos * n_neg)


class _TrainAwareMetric(Metric):
    """Base for metrics that need the train log (Coverage, Novelty, Surprisal)."""

    def __call__(
        self,
        recommendations: MetricsDataFrameLike,
        train: Optional[MetricsDataFrameLike] = None,
        ground_truth: Optional[MetricsDataFrameLike] = None,
    ) -> MetricsReturnType:
        if train is None:
            raise ValueError(f"{self.__name__} requires the train interactions")
        recs = self._recs_to_dict(recommendations)
        train_dict = self._gt_to_dict(train)
        return self._compute_with_train(recs, train_dict)

    def _compute_with_train(self, recs: Dict, train: Dict) -> MetricsReturnType:  # pragma: no cover
        raise NotImplementedError


class Coverage(_TrainAwareMetric):
    """Share of catalog items appearing in anyone's top-k
    (reference coverage.py)."""

    def _compute_with_train(self, recs: Dict, train: Dict) -> MetricsReturnType:
        catalog = set()
        for items in train.values():
            catalog.update(items)
        result = {}
        for k in self.topk:
            recommended = set()
            for pred in recs.values():
                recommended.update(pred[:k])
            result[f"{self.__name__}@{k}"] = len(recommended & catalog) / len(catalog) if catalog else 0.0
        return result


class Novelty(_TrainAwareMetric):
    """Per-user share of recommended items the user has NOT interacted with in
    train (reference novelty.py)."""

    def _compute_with_train(self, recs: Dict, train: Dict) -> MetricsReturnType:
        result = {}
        queries = list(train.keys())  # reference iterates TRAIN users
        for k in self.topk:
            values = []
            for q in queries:
                pred = recs.get(q, [])
                seen = set(train.get(q, []))
                topk_items = pred[:k]
                if not topk_items or not seen:
                    values.append(1.0)  # reference novelty.py:144: vacuously novel
                    continue
                values.append(sum(1 for p in topk_items if p not in seen) / len(topk_items))
            result.update(self._format_result(k, np.asarray(values, dtype=np.float64), queries))
        return result


class Surprisal(_TrainAwareMetric):
    """Mean self-information -log2(pop_share) of recommended items, normalized
    by log2(n_users) (reference surprisal.py)."""

    def _compute_with_train(self, recs: Dict, train: Dict) -> MetricsReturnType:
        item_users: Dict = {}
        for q, items in train.items():
            for i in set(items):
                item_users[i] = item_users.get(i, 0) + 1
        n_users = max(1, len(train))
        max_info = np.log2(n_users) if n_users > 1 else 1.0
        result = {}
        for k in self.topk:
            values = []
            for q, pred in recs.items():
                # reference surprisal.py:188 divides the weight SUM by K
                # (short lists lower the score); unseen items weigh 1.0
                total = 0.0
                for p in pred[:k]:
                    share = item_users.get(p, 0) / n_users
                    info = -np.log2(share) if share > 0 else max_info
                    total += info / max_info
                values.append(total / k if pred else 0.0)
            result.update(
                self._format_result(k, np.asarray(values, dtype=np.float64), list(recs.keys()))
            )
        return result


class Unexpectedness(Metric):
    """Share of recommended items not predicted by a baseline model
    (reference unexpectedness.py).  Call with (recommendations,
    base_recommendations)."""

    def __call__(
        self,
        recommendations: MetricsDataFrameLike,
        base_recommendations: MetricsDataFrameLike,
    ) -> MetricsReturnType:
        recs = self._recs_to_dict(recommendations)
        base = self._recs_to_dict(base_recommendations)
        result = {}
        for k in self.topk:
            values = []
            for q, pred in recs.items():
                base_set = set(base.get(q, [])[:k])
                topk_items = pred[:k]
                if not topk_items:
                    values.append(0.0)
                    continue
                # reference unexpectedness.py:156 divides by K (not by the
                # number of produced recs): short lists count as unexpected
                values.append(1.0 - len(set(topk_items) & base_set) / k)
            result.update(
                self._format_result(k, np.asarray(values, dtype=np.float64), list(recs.keys()))
            )
        return result


class CategoricalDiversity(Metric):
    """Mean share of distinct item categories in top-k
    (reference categorical_diversity.py).  Recommendations must carry a
    category column: call with a DataFrame [query, category, rating]."""

    def __init__(self, topk, query_column="query_id", category_column="category_id", rating_column="rating", mode=None):
        super().__init__(
            topk=topk,
            query_column=query_column,
            item_column=category_column,
            rating_column=rating_column,
            mode=mode,
        )
        self.category_column = category_column

    def __call__(self, recommendations: MetricsDataFrameLike) -> MetricsReturnType:
        recs = self._recs_to_dict(recommendations)
        result = {}
        for k in self.topk:
            values = []
            for q, cats in recs.items():
                topk_cats = cats[:k]
                if not topk_cats:
                    values.append(0.0)
                    continue
                # reference categorical_diversity.py divides by K: fewer
                # than K recommendations lowers diversity
                values.append(len(set(topk_cats)) / k)
            result[f"{self.__name__}@{k}"] = self._mode.cpu(np.asarray(values, dtype=np.float64))
        return result
