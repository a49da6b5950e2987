from .callback.metrics_callback import ComputeMetricsCallback
from .callback.predictions_callback import (
    HiddenStatesCallback,
    PandasTopItemsCallback,
    PolarsTopItemsCallback,
    QueryEmbeddingsPredictionCallback,
    SparkTopItemsCallback,
    TopItemsCallbackBase,
    TorchTopItemsCallback,
)
from .module import LightningModule
from .optimizer import LambdaLRSchedulerFactory, LRSchedulerFactory, OptimizerFactory
from .postprocessor.seen_items import BasePostProcessor, SampleItemsFilter, SeenItemsFilter

__all__ = [
    "ComputeMetricsCallback",
    "HiddenStatesCallback",
    "PandasTopItemsCallback",
    "PolarsTopItemsCallback",
    "QueryEmbeddingsPredictionCallback",
    "SparkTopItemsCallback",
    "TopItemsCallbackBase",
    "TorchTopItemsCallback",
    "LightningModule",
    "LambdaLRSchedulerFactory",
    "LRSchedulerFactory",
    "OptimizerFactory",
    "BasePostProcessor",
    "SampleItemsFilter",
    "SeenItemsFilter",
]
