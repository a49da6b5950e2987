"""Legacy-surface metric classes (reference experimental/metrics/*.py, one
file per metric there): each delegates to the replay_amd.metrics
implementation through the legacy [user_idx, item_idx, relevance] call
interface of ``base_metric.Metric``."""

from .base_metric import Metric


class HitRate(Metric):
    """At least one hit in top-k (reference hitrate.py:4)."""


class Precision(Metric):
    """Share of relevant items in top-k (reference precision.py)."""


class Recall(Metric):
    """Share of the ground truth recovered in top-k (reference recall.py)."""


class MAP(Metric):
    """Mean average precision (reference map.py)."""


class MRR(Metric):
    """Mean reciprocal rank (reference mrr.py)."""


class NDCG(Metric):
    """Normalized discounted cumulative gain (reference ndcg.py)."""


class RocAuc(Metric):
    """Per-user ROC AUC (reference rocauc.py)."""


class Coverage(Metric):
    """Share of the catalog recommended (reference coverage.py).  Needs the
    train log for the catalog: pass it as ground_truth."""

    _main_metric_name = "Coverage"


class Surprisal(Metric):
    """Self-information of recommended items (reference surprisal.py)."""

    _main_metric_name = "Surprisal"


class Unexpectedness(Metric):
    """Share of recommendations absent from a baseline model's list
    (reference unexpectedness.py)."""

    _main_metric_name = "Unexpectedness"
