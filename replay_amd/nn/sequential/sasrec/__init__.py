from .agg import PositionAwareAggregator
from .model import SasRec, SasRecBody
from .transformer import SasRecTransformerBlock, SasRecTransformerLayer

__all__ = [
    "PositionAwareAggregator",
    "SasRec",
    "SasRecBody",
    "SasRecTransformerBlock",
    "SasRecTransformerLayer",
]
