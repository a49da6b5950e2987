"""Legacy experimental-metrics surface (reference replay/experimental/
metrics): the classic metric set under the old callable interface
``metric(recs, ground_truth, k)`` over ``[user_idx, item_idx, relevance]``
frames, plus NCIS (Normalized Capped Importance Sampling) counterfactual
weighting (arxiv.org/abs/1801.07030) for off-policy evaluation."""

from .base_metric import Metric, NCISMetric
from .metrics import MAP, MRR, NDCG, Coverage, HitRate, Precision, Recall, RocAuc, Surprisal, Unexpectedness
from .ncis_precision import NCISPrecision

__all__ = [
    "Metric",
    "NCISMetric",
    "NCISPrecision",
    "Coverage",
    "HitRate",
    "MAP",
    "MRR",
    "NDCG",
    "Precision",
    "Recall",
    "RocAuc",
    "Surprisal",
    "Unexpectedness",
]
