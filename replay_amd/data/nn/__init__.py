from .parquet import ParquetDataset, ParquetModule
from .schema import MutableTensorMap, TensorFeatureInfo, TensorFeatureSource, TensorMap, TensorSchema
from .sequence_tokenizer import SequenceTokenizer
from .sequential_dataset import PandasSequentialDataset, PolarsSequentialDataset, SequentialDataset
from .torch_sequential_dataset import (
    DEFAULT_GROUND_TRUTH_PADDING_VALUE,
    DEFAULT_TRAIN_PADDING_VALUE,
    GROUND_TRUTH_PAD,
    TRAIN_PAD,
    TorchSequentialBatch,
    TorchSequentialDataset,
    TorchSequentialValidationBatch,
    TorchSequentialValidationDataset,
)

__all__ = [
    "DEFAULT_GROUND_TRUTH_PADDING_VALUE",
    "DEFAULT_TRAIN_PADDING_VALUE",
    "GROUND_TRUTH_PAD",
    "TRAIN_PAD",
    "MutableTensorMap",
    "PandasSequentialDataset",
    "ParquetDataset",
    "ParquetModule",
    "PolarsSequentialDataset",
    "SequenceTokenizer",
    "SequentialDataset",
    "TensorFeatureInfo",
    "TensorFeatureSource",
    "TensorMap",
    "TensorSchema",
    "TorchSequentialBatch",
    "TorchSequentialDataset",
    "TorchSequentialValidationBatch",
    "TorchSequentialValidationDataset",
]
