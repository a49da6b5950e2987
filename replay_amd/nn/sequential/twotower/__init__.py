from .model import ItemTower, QueryTower, TwoTower, TwoTowerBody, TwoTowerHead
from .reader import FeaturesReader, FeaturesReaderProtocol

__all__ = [
    "ItemTower", "QueryTower", "TwoTower", "TwoTowerBody", "TwoTowerHead",
    "FeaturesReader", "FeaturesReaderProtocol",
]
