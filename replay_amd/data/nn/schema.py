"""Tensor feature metadata for sequence models.

Parity with reference replay/data/nn/schema.py (TensorFeatureSource:13,
TensorFeatureInfo:56, TensorSchema:242): per-feature ``is_seq``,
``cardinality``, ``padding_value``, ``embedding_dim``, ``tensor_dim`` and
source bookkeeping.
"""

from __future__ import annotations

from typing import Mapping, Dict, Iterator, List, Optional, Sequence, Union

from replay_amd.data.schema import FeatureHint, FeatureSource, FeatureType


TensorMap = Mapping[str, "torch.Tensor"]
MutableTensorMap = Dict[str, "torch.Tensor"]


class TensorFeatureSource:
    """Where a tensor feature came from (frame + column)."""

    def __init__(self, source: FeatureSource, column: str, index: Optional[int] = None) -> None:
        self._source = source
        self._column = column
        self._index = index

    @property
    def source(self) -> FeatureSource:
        return self._source

    @property
    def column(self) -> str:
        return self._column

    @property
    def index(self) -> Optional[int]:
        return self._index

    def __repr__(self) -> str:  # pragma: no cover
        return f"TensorFeatureSource({self._source}, {self._column!r})"


class TensorFeatureInfo:
    """Metadata of one tensor feature."""

    def __init__(
        self,
        name: str,
        feature_type: FeatureType,
        is_seq: bool = False,
        feature_hint: Optional[FeatureHint] = None,
        feature_sources: Optional[List[TensorFeatureSource]] = None,
        cardinality: Optional[int] = None,
        embedding_dim: Optional[int] = None,
        tensor_dim: Optional[int] = None,
        padding_value: int = 0,
    ) -> None:
        self._name = name
        if feature_type not in FeatureType:
            raise ValueError(f"Unknown feature type {feature_type}")
        self._feature_type = feature_type
        self._is_seq = is_seq
        self._feature_hint = feature_hint
        self._feature_sources = feature_sources or []
        self._padding_value = padding_value

        numerical = feature_type in (FeatureType.NUMERICAL, FeatureType.NUMERICAL_LIST)
        if numerical and cardinality is not None:
            raise ValueError("cardinality is only for categorical features")
        if not numerical and tensor_dim is not None:
            raise ValueError("tensor_dim is only for numerical features")
        self._cardinality = cardinality
        self._embedding_dim = embedding_dim
        self._tensor_dim = tensor_dim

    # -- accessors -------------------------------------------------------------
    @property
    def name(self) -> str:
        return self._name

    @property
    def feature_type(self) -> FeatureType:
        return self._feature_type

    @property
    def is_seq(self) -> bool:
        return self._is_seq

    @property
    def feature_hint(self) -> Optional[FeatureHint]:
        return self._feature_hint

    @property
    def feature_sources(self) -> List[TensorFeatureSource]:
        return list(self._feature_sources)

    @property
    def feature_source(self) -> Optional[TensorFeatureSource]:
        return self._feature_sources[0] if self._feature_sources else None

    @property
    def is_cat(self) -> bool:
        return self._feature_type in (FeatureType.CATEGORICAL, FeatureType.CATEGORICAL_LIST)

    @property
    def is_num(self) -> bool:
        return not self.is_cat

    @property
    def is_list(self) -> bool:
        return self._feature_type in (FeatureType.CATEGORICAL_LIST, FeatureType.NUMERICAL_LIST)

    @property
    def cardinality(self) -> Optional[int]:
        if not self.is_cat:
            raise RuntimeError(f"cardinality is undefined for numerical feature {self._name}")
        return self._cardinality

    @property
    def padding_value(self) -> int:
        return self._padding_value

    @property
    def embedding_dim(self) -> Optional[int]:
        return self._embedding_dim

    @property
    def tensor_dim(self) -> Optional[int]:
        if self.is_cat:
            raise RuntimeError(f"tensor_dim is undefined for categorical feature {self._name}")
        return self._tensor_dim

    def _set_cardinality(self, cardinality: int) -> None:
        self._cardinality = cardinality

    def _set_embedding_dim(self, dim: int) -> None:
        self._embedding_dim = dim

    def __repr__(self) -> str:  # pragma: no cover
        return f"TensorFeatureInfo({self._name!r}, {self._feature_type.name}, seq={self._is_seq})"


class TensorSchema:
    """Ordered mapping name -> TensorFeatureInfo with convenience selectors."""

    def __init__(self, features_list: Union[Sequence[TensorFeatureInfo], TensorFeatureInfo]) -> None:
        if isinstance(features_list, TensorFeatureInfo):
            features_list = [features_list]
        self._features: Dict[str, TensorFeatureInfo] = {}
        for f in features_list:
            if f.name in self._features:
                raise ValueError(f"Duplicate tensor feature {f.name}")
            self._features[f.name] = f

    # -- mapping interface -----------------------------------------------------
    def __getitem__(self, name: str) -> TensorFeatureInfo:
        return self._features[name]

    def __contains__(self, name: str) -> bool:
        return name in self._features

    def __iter__(self) -> Iterator[str]:
        return iter(self._features)

    def __len__(self) -> int:
        return len(self._features)

    def __add__(self, other: "TensorSchema") -> "TensorSchema":
        return TensorSchema(list(self._features.values()) + list(other._features.values()))

    def keys(self):
        return self._features.keys()

    def values(self):
        return self._features.values()

    def items(self):
        return self._features.items()

    def get(self, name: str, default=None):
        return self._features.get(name, default)

    def subset(self, names: Sequence[str]) -> "TensorSchema":
        return TensorSchema([self._features[n] for n in names if n in self._features])

    def filter(self, name: Optional[str] = None, feature_hint: Optional[FeatureHint] = None, is_seq: Optional[bool] = None, feature_type: Optional[FeatureType] = None) -> "TensorSchema":
        feats = list(self._features.values())
        if name is not None:
            feats = [f for f in feats if f.name == name]
        if feature_hint is not None:
            feats = [f for f in feats if f.feature_hint == feature_hint]
        if is_seq is not None:
            feats = [f for f in feats if f.is_seq == is_seq]
        if feature_type is not None:
            feats = [f for f in feats if f.feature_type == feature_type]
        return TensorSchema(feats)

    # -- selectors --------------------------------------------------------------
    @property
    def all_features(self) -> List[TensorFeatureInfo]:
        return list(self._features.values())

    @property
    def categorical_features(self) -> "TensorSchema":
        return TensorSchema([f for f in self._features.values() if f.is_cat])

    @property
    def numerical_features(self) -> "TensorSchema":
        return TensorSchema([f for f in self._features.values() if f.is_num])

    @property
    def sequential_features(self) -> "TensorSchema":
        return TensorSchema([f for f in self._features.values() if f.is_seq])

    def _hinted(self, hint: FeatureHint) -> Optional[TensorFeatureInfo]:
        for f in self._features.values():
            if f.feature_hint == hint:
                return f
        return None

    @property
    def item_id_feature_name(self) -> Optional[str]:
        f = self._hinted(FeatureHint.ITEM_ID)
        return f.name if f else None

    @property
    def query_id_feature_name(self) -> Optional[str]:
        f = self._hinted(FeatureHint.QUERY_ID)
        return f.name if f else None

    @property
    def item_id_features(self) -> "TensorSchema":
        return self.filter(feature_hint=FeatureHint.ITEM_ID)

    @property
    def timestamp_feature_name(self) -> Optional[str]:
        f = self._hinted(FeatureHint.TIMESTAMP)
        return f.name if f else None

    @property
    def rating_feature_name(self) -> Optional[str]:
        f = self._hinted(FeatureHint.RATING)
        return f.name if f else None

    # -- serialization -----------------------------------------------------------
    def to_dict(self) -> List[Dict]:
        out = []
        for f in self._features.values():
            out.append(
                {
                    "name": f.name,
                    "feature_type": f.feature_type.value,
                    "is_seq": f.is_seq,
                    "feature_hint": f.feature_hint.value if f.feature_hint else None,
                    "cardinality": f._cardinality,
                    "embedding_dim": f._embedding_dim,
                    "tensor_dim": f._tensor_dim,
                    "padding_value": f.padding_value,
                    "sources": [
                        {"source": s.source.value, "column": s.column, "index": s.index}
                        for s in f.feature_sources
                    ],
                }
            )
        return out

    @classmethod
    def from_dict(cls, data: List[Dict]) -> "TensorSchema":
        feats = []
        for d in data:
            ftype = FeatureType(d["feature_type"])
            numerical = ftype in (FeatureType.NUMERICAL, FeatureType.NUMERICAL_LIST)
            feats.append(
                TensorFeatureInfo(
                    name=d["name"],
                    feature_type=ftype,
                    is_seq=d["is_seq"],
                    feature_hint=FeatureHint(d["feature_hint"]) if d.get("feature_hint") else None,
                    cardinality=None if numerical else d.get("cardinality"),
                    embedding_dim=d.get("embedding_dim"),
                    tensor_dim=d.get("tensor_dim") if numerical else None,
                    padding_value=d.get("padding_value", 0),
                    feature_sources=[
                        TensorFeatureSource(FeatureSource(s["source"]), s["column"], s.get("index"))
                        for s in d.get("sources", [])
                    ],
                )
            )
        return cls(feats)
