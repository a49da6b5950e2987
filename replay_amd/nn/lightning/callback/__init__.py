from .metrics_callback import ComputeMetricsCallback
from .predictions_callback import (
    HiddenStatesCallback,
    PandasTopItemsCallback,
    PolarsTopItemsCallback,
    QueryEmbeddingsPredictionCallback,
    SparkTopItemsCallback,
    TopItemsCallbackBase,
    TorchTopItemsCallback,
)

__all__ = [
    "ComputeMetricsCallback",
    "HiddenStatesCallback",
    "PandasTopItemsCallback",
    "PolarsTopItemsCallback",
    "QueryEmbeddingsPredictionCallback",
    "SparkTopItemsCallback",
    "TopItemsCallbackBase",
    "TorchTopItemsCallback",
]
