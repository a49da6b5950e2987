"""Fluent TensorSchema builder (reference experimental/nn/data/
schema_builder.py:5) — the legacy chained-call surface over
replay_amd.data.nn.TensorSchema."""

from __future__ import annotations

from typing import Dict, List, Optional

from replay_amd.data.nn.schema import TensorFeatureInfo, TensorFeatureSource, TensorSchema
from replay_amd.data.schema import FeatureHint, FeatureType


class TensorSchemaBuilder:
    """Chain ``.categorical(...)/.numerical(...)`` calls, then ``.build()``."""

    def __init__(self) -> None:
        self._features: Dict[str, TensorFeatureInfo] = {}

    def categorical(
        self,
        name: str,
        cardinality: int,
        is_seq: bool = False,
        feature_source: Optional[TensorFeatureSource] = None,
        feature_hint: Optional[FeatureHint] = None,
        embedding_dim: Optional[int] = None,
        padding_value: int = 0,
    ) -> "TensorSchemaBuilder":
        self._features[name] = TensorFeatureInfo(
            name=name,
            feature_type=FeatureType.CATEGORICAL,
            is_seq=is_seq,
            feature_sources=[feature_source] if feature_source else None,
            feature_hint=feature_hint,
            cardinality=cardinality,
            embedding_dim=embedding_dim,
            padding_value=padding_value,
        )
        return self

    def numerical(
        self,
        name: str,
        tensor_dim: int,
        is_seq: bool = False,
        feature_sources: Optional[List[TensorFeatureSource]] = None,
        feature_hint: Optional[FeatureHint] = None,
        padding_value: int = 0,
    ) -> "TensorSchemaBuilder":
        self._features[name] = TensorFeatureInfo(
            name=name,
            feature_type=FeatureType.NUMERICAL,
            is_seq=is_seq,
            feature_sources=feature_sources,
            feature_hint=feature_hint,
            tensor_dim=tensor_dim,
            padding_value=padding_value,
        )
        return self

    def categorical_list(
        self,
        name: str,
        cardinality: int,
        is_seq: bool = False,
        feature_source: Optional[TensorFeatureSource] = None,
        feature_hint: Optional[FeatureHint] = None,
        embedding_dim: Optional[int] = None,
        padding_value: int = 0,
    ) -> "TensorSchemaBuilder":
        self._features[name] = TensorFeatureInfo(
            name=name,
            feature_type=FeatureType.CATEGORICAL_LIST,
            is_seq=is_seq,
            feature_sources=[feature_source] if feature_source else None,
            feature_hint=feature_hint,
            cardinality=cardinality,
            embedding_dim=embedding_dim,
            padding_value=padding_value,
        )
        return self

    def numerical_list(
        self,
        name: str,
        tensor_dim: int,
        is_seq: bool = False,
        feature_sources: Optional[List[TensorFeatureSource]] = None,
        feature_hint: Optional[FeatureHint] = None,
        padding_value: int = 0,
    ) -> "TensorSchemaBuilder":
        self._features[name] = TensorFeatureInfo(
            name=name,
            feature_type=FeatureType.NUMERICAL_LIST,
            is_seq=is_seq,
            feature_sources=feature_sources,
            feature_hint=feature_hint,
            tensor_dim=tensor_dim,
            padding_value=padding_value,
        )
        return self

    def build(self) -> TensorSchema:
        return TensorSchema(list(self._features.values()))
