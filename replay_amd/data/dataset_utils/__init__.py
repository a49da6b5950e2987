from .dataset_label_encoder import DatasetLabelEncoder

__all__ = ["DatasetLabelEncoder"]
