from .schema import TensorFeatureInfo, TensorFeatureSource, TensorSchema
from .sequence_tokenizer import SequenceTokenizer
from .sequential_dataset import PandasSequentialDataset, SequentialDataset
from .torch_sequential_dataset import (
    GROUND_TRUTH_PAD,
    TRAIN_PAD,
    TorchSequentialDataset,
    TorchSequentialValidationDataset,
)

__all__ = [
    "TensorFeatureInfo",
    "TensorFeatureSource",
    "TensorSchema",
    "SequenceTokenizer",
    "PandasSequentialDataset",
    "SequentialDataset",
    "GROUND_TRUTH_PAD",
    "TRAIN_PAD",
    "TorchSequentialDataset",
    "TorchSequentialValidationDataset",
]
