from .bert4rec import Bert4Rec, Bert4RecBody
from .sasrec import (
    DiffTransformerBlock,
    DiffTransformerLayer,
    PositionAwareAggregator,
    SasRec,
    SasRecBody,
    SasRecTransformerBlock,
    SasRecTransformerLayer,
)
from .twotower import FeaturesReader, FeaturesReaderProtocol, ItemTower, QueryTower, TwoTower, TwoTowerBody

__all__ = [
    "Bert4Rec",
    "Bert4RecBody",
    "DiffTransformerBlock",
    "DiffTransformerLayer",
    "PositionAwareAggregator",
    "SasRec",
    "SasRecBody",
    "SasRecTransformerBlock",
    "SasRecTransformerLayer",
    "FeaturesReader",
    "FeaturesReaderProtocol",
    "ItemTower",
    "QueryTower",
    "TwoTower",
    "TwoTowerBody",
]
