from .model import ItemTower, QueryTower, TwoTower, TwoTowerBody, TwoTowerHead
from .reader import FeaturesReader

__all__ = ["ItemTower", "QueryTower", "TwoTower", "TwoTowerBody", "TwoTowerHead", "FeaturesReader"]
