from .dataset import Dataset
from .frame_schema import get_schema
from .schema import FeatureHint, FeatureInfo, FeatureSchema, FeatureSource, FeatureType

__all__ = [
    "Dataset",
    "FeatureHint",
    "FeatureInfo",
    "FeatureSchema",
    "FeatureSource",
    "FeatureType",
    "get_schema",
]
