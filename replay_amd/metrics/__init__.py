from .base_metric import (
    CalculationDescriptor,
    ConfidenceInterval,
    Mean,
    Median,
    Metric,
    PerUser,
)
from .metrics import (
    MAP,
    MRR,
    NDCG,
    CategoricalDiversity,
    Coverage,
    HitRate,
    Novelty,
    Precision,
    Recall,
    RocAuc,
    Surprisal,
    Unexpectedness,
)
from .offline_metrics import Experiment, OfflineMetrics
from .torch_metrics_builder import TorchMetricsBuilder

__all__ = [
    "CalculationDescriptor",
    "ConfidenceInterval",
    "Mean",
    "Median",
    "Metric",
    "PerUser",
    "MAP",
    "MRR",
    "NDCG",
    "CategoricalDiversity",
    "Coverage",
    "HitRate",
    "Novelty",
    "Precision",
    "Recall",
    "RocAuc",
    "Surprisal",
    "Unexpectedness",
    "Experiment",
    "OfflineMetrics",
    "TorchMetricsBuilder",
]
