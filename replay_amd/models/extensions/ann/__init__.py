from .ann_mixin import ANNMixin
from .index import BruteForceIndex, IndexParams

__all__ = ["ANNMixin", "BruteForceIndex", "IndexParams"]
