from .dataset import Dataset
from .schema import FeatureHint, FeatureInfo, FeatureSchema, FeatureSource, FeatureType

__all__ = [
    "Dataset",
    "FeatureHint",
    "FeatureInfo",
    "FeatureSchema",
    "FeatureSource",
    "FeatureType",
]
