from .agg import PositionAwareAggregator
from .diff_transformer import DiffTransformerBlock, DiffTransformerLayer
from .model import SasRec, SasRecBody
from .transformer import SasRecTransformerBlock, SasRecTransformerLayer

__all__ = [
    "DiffTransformerBlock",
    "DiffTransformerLayer",
    "PositionAwareAggregator",
    "SasRec",
    "SasRecBody",
    "SasRecTransformerBlock",
    "SasRecTransformerLayer",
]
