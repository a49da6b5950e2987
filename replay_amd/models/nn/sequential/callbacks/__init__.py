"""Legacy callbacks namespace (reference replay/models/nn/sequential/
callbacks/prediction_callbacks.py, validation_callback.py) — the MI355X build
keeps one callback implementation; these are the legacy import paths."""

from replay_amd.nn.lightning.callback.metrics_callback import (
    ComputeMetricsCallback as ValidationMetricsCallback,
)
from replay_amd.nn.lightning.callback.predictions_callback import (
    TopItemsCallbackBase as BasePredictionCallback,
    PandasTopItemsCallback as PandasPredictionCallback,
    PolarsTopItemsCallback as PolarsPredictionCallback,
    QueryEmbeddingsPredictionCallback,
    SparkTopItemsCallback as SparkPredictionCallback,
    TorchTopItemsCallback as TorchPredictionCallback,
)

__all__ = [
    "BasePredictionCallback",
    "ValidationMetricsCallback",
    "PandasPredictionCallback",
    "PolarsPredictionCallback",
    "QueryEmbeddingsPredictionCallback",
    "SparkPredictionCallback",
    "TorchPredictionCallback",
]
