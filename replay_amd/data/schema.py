"""Typed feature metadata for interaction datasets.

API parity with the reference schema layer (reference: replay/data/schema.py:
FeatureType:5, FeatureSource:14, FeatureHint:22, FeatureInfo:31,
FeatureSchema:119) re-implemented for the pandas/Arrow-first MI355X build.
"""

from __future__ import annotations

from enum import Enum
from typing import Callable, Dict, Iterable, Iterator, List, Optional, Sequence, Union


class FeatureType(Enum):
    """Type of a feature column."""

    CATEGORICAL = "categorical"
    NUMERICAL = "numerical"
    CATEGORICAL_LIST = "categorical_list"
    NUMERICAL_LIST = "numerical_list"


class FeatureSource(Enum):
    """Which frame a feature column lives in."""

    ITEM_FEATURES = "item_features"
    QUERY_FEATURES = "query_features"
    INTERACTIONS = "interactions"


class FeatureHint(Enum):
    """Special role of a feature column."""

    ITEM_ID = "item_id"
    QUERY_ID = "query_id"
    RATING = "rating"
    TIMESTAMP = "timestamp"


class FeatureInfo:
    """Metadata for a single feature column.

    Cardinality is lazy: it is computed by a callback installed by the owning
    ``Dataset`` (reference: replay/data/schema.py:105-110).
    """

    def __init__(
        self,
        column: str,
        feature_type: FeatureType,
        feature_hint: Optional[FeatureHint] = None,
        feature_source: Optional[FeatureSource] = None,
        cardinality: Optional[int] = None,
    ) -> None:
        self._column = column
        if not isinstance(feature_type, FeatureType):
            raise ValueError(f"Unknown feature type: {feature_type}")
        self._feature_type = feature_type
        self._feature_hint = feature_hint
        self._feature_source = feature_source
        if feature_type in (FeatureType.NUMERICAL, FeatureType.NUMERICAL_LIST) and cardinality is not None:
            raise ValueError(f"Cardinality is only defined for categorical features, got {column}")
        self._cardinality = cardinality
        self._cardinality_callback: Optional[Callable[[str], int]] = None

    @property
    def column(self) -> str:
        return self._column

    @property
    def feature_type(self) -> FeatureType:
        return self._feature_type

    @property
    def feature_hint(self) -> Optional[FeatureHint]:
        return self._feature_hint

    @property
    def feature_source(self) -> Optional[FeatureSource]:
        return self._feature_source

    def _set_feature_source(self, source: FeatureSource) -> None:
        self._feature_source = source

    def _set_cardinality_callback(self, callback: Callable[[str], int]) -> None:
        self._cardinality_callback = callback

    @property
    def cardinality(self) -> Optional[int]:
        if self._feature_type not in (FeatureType.CATEGORICAL, FeatureType.CATEGORICAL_LIST):
            raise RuntimeError(f"Cardinality is not defined for {self._feature_type.name} feature {self._column}")
        if self._cardinality is None and self._cardinality_callback is not None:
            self._cardinality = self._cardinality_callback(self._column)
        return self._cardinality

    def reset_cardinality(self) -> None:
        self._cardinality = None

    def __repr__(self) -> str:  # pragma: no cover
        return (
            f"FeatureInfo(column={self._column!r}, type={self._feature_type.name}, "
            f"hint={self._feature_hint}, source={self._feature_source})"
        )

    def copy(self) -> "FeatureInfo":
        cardinality = None
        if self._feature_type in (FeatureType.CATEGORICAL, FeatureType.CATEGORICAL_LIST):
            cardinality = self._cardinality
        return FeatureInfo(
            column=self._column,
            feature_type=self._feature_type,
            feature_hint=self._feature_hint,
            feature_source=self._feature_source,
            cardinality=cardinality,
        )


class FeatureSchema:
    """Immutable ordered collection of :class:`FeatureInfo`.

    Mirrors reference FeatureSchema (replay/data/schema.py:119) including the
    ``subset``/``filter``/``drop`` API and the uniqueness checks on
    query/item/timestamp/rating hints (reference schema.py:433-465).
    """

    def __init__(self, features_list: Union[Sequence[FeatureInfo], FeatureInfo]) -> None:
        if isinstance(features_list, FeatureInfo):
            features_list = [features_list]
        self._features: List[FeatureInfo] = list(features_list)
        self._check_unique_columns()
        self._check_unique_hints()

    # -- mapping-like interface ------------------------------------------------
    def __iter__(self) -> Iterator[FeatureInfo]:
        return iter(self._features)

    def __len__(self) -> int:
        return len(self._features)

    def __getitem__(self, column: str) -> FeatureInfo:
        for feature in self._features:
            if feature.column == column:
                return feature
        raise KeyError(column)

    def __contains__(self, column: str) -> bool:
        return any(f.column == column for f in self._features)

    def get(self, column: str, default: Optional[FeatureInfo] = None) -> Optional[FeatureInfo]:
        try:
            return self[column]
        except KeyError:
            return default

    def keys(self) -> List[str]:
        return [f.column for f in self._features]

    def values(self) -> List[FeatureInfo]:
        return list(self._features)

    def items(self) -> List[tuple]:
        return [(f.column, f) for f in self._features]

    def __add__(self, other: "FeatureSchema") -> "FeatureSchema":
        return FeatureSchema(list(self._features) + list(other._features))

    def copy(self) -> "FeatureSchema":
        return FeatureSchema([f.copy() for f in self._features])

    # -- checks ----------------------------------------------------------------
    def _check_unique_columns(self) -> None:
        names = [f.column for f in self._features]
        if len(names) != len(set(names)):
            dupes = {n for n in names if names.count(n) > 1}
            raise ValueError(f"Duplicate feature columns in schema: {sorted(dupes)}")

    def _check_unique_hints(self) -> None:
        for hint in (FeatureHint.QUERY_ID, FeatureHint.ITEM_ID, FeatureHint.RATING, FeatureHint.TIMESTAMP):
            hinted = [f for f in self._features if f.feature_hint == hint]
            if len(hinted) > 1:
                raise ValueError(f"Multiple columns with hint {hint}: {[f.column for f in hinted]}")

    # -- selection -------------------------------------------------------------
    def subset(self, features_to_keep: Iterable[str]) -> "FeatureSchema":
        keep = set(features_to_keep)
        return FeatureSchema([f.copy() for f in self._features if f.column in keep])

    def filter(
        self,
        column: Optional[str] = None,
        feature_hint: Optional[FeatureHint] = None,
        feature_source: Optional[FeatureSource] = None,
        feature_type: Optional[FeatureType] = None,
    ) -> "FeatureSchema":
        result = list(self._features)
        if column is not None:
            result = [f for f in result if f.column == column]
        if feature_hint is not None:
            result = [f for f in result if f.feature_hint == feature_hint]
        if feature_source is not None:
            result = [f for f in result if f.feature_source == feature_source]
        if feature_type is not None:
            result = [f for f in result if f.feature_type == feature_type]
        return FeatureSchema([f.copy() for f in result])

    def drop(
        self,
        column: Optional[str] = None,
        feature_hint: Optional[FeatureHint] = None,
        feature_source: Optional[FeatureSource] = None,
        feature_type: Optional[FeatureType] = None,
    ) -> "FeatureSchema":
        result = list(self._features)
        if column is not None:
            result = [f for f in result if f.column != column]
        if feature_hint is not None:
            result = [f for f in result if f.feature_hint != feature_hint]
        if feature_source is not None:
            result = [f for f in result if f.feature_source != feature_source]
        if feature_type is not None:
            result = [f for f in result if f.feature_type != feature_type]
        return FeatureSchema([f.copy() for f in result])

    # -- convenience accessors (parity with reference properties) --------------
    @property
    def all_features(self) -> List[FeatureInfo]:
        return list(self._features)

    @property
    def columns(self) -> List[str]:
        return self.keys()

    def _hinted(self, hint: FeatureHint) -> Optional[FeatureInfo]:
        for f in self._features:
            if f.feature_hint == hint:
                return f
        return None

    @property
    def query_id_feature(self) -> Optional[FeatureInfo]:
        return self._hinted(FeatureHint.QUERY_ID)

    @property
    def item_id_feature(self) -> Optional[FeatureInfo]:
        return self._hinted(FeatureHint.ITEM_ID)

    @property
    def interactions_rating_feature(self) -> Optional[FeatureInfo]:
        return self._hinted(FeatureHint.RATING)

    @property
    def interactions_timestamp_feature(self) -> Optional[FeatureInfo]:
        return self._hinted(FeatureHint.TIMESTAMP)

    @property
    def query_id_column(self) -> Optional[str]:
        f = self.query_id_feature
        return f.column if f is not None else None

    @property
    def item_id_column(self) -> Optional[str]:
        f = self.item_id_feature
        return f.column if f is not None else None

    @property
    def interactions_rating_column(self) -> Optional[str]:
        f = self.interactions_rating_feature
        return f.column if f is not None else None

    @property
    def interactions_timestamp_column(self) -> Optional[str]:
        f = self.interactions_timestamp_feature
        return f.column if f is not None else None

    @property
    def categorical_features(self) -> "FeatureSchema":
        return self.filter(feature_type=FeatureType.CATEGORICAL)

    @property
    def numerical_features(self) -> "FeatureSchema":
        return self.filter(feature_type=FeatureType.NUMERICAL)

    @property
    def item_features(self) -> "FeatureSchema":
        return self.filter(feature_source=FeatureSource.ITEM_FEATURES)

    @property
    def query_features(self) -> "FeatureSchema":
        return self.filter(feature_source=FeatureSource.QUERY_FEATURES)

    @property
    def interaction_features(self) -> "FeatureSchema":
        return FeatureSchema(
            [
                f.copy()
                for f in self._features
                if f.feature_source == FeatureSource.INTERACTIONS
                and f.feature_hint not in (FeatureHint.QUERY_ID, FeatureHint.ITEM_ID)
            ]
        )

    @property
    def interactions_features(self) -> "FeatureSchema":
        # alias used by some reference call sites
        return self.interaction_features

    # -- serialization ---------------------------------------------------------
    def to_dict(self) -> List[Dict]:
        out = []
        for f in self._features:
            cardinality = None
            if f.feature_type in (FeatureType.CATEGORICAL, FeatureType.CATEGORICAL_LIST):
                cardinality = f._cardinality
            out.append(
                {
                    "column": f.column,
                    "feature_type": f.feature_type.value,
                    "feature_hint": f.feature_hint.value if f.feature_hint else None,
                    "feature_source": f.feature_source.value if f.feature_source else None,
                    "cardinality": cardinality,
                }
            )
        return out

    @classmethod
    def from_dict(cls, data: List[Dict]) -> "FeatureSchema":
        features = []
        for d in data:
            features.append(
                FeatureInfo(
                    column=d["column"],
                    feature_type=FeatureType(d["feature_type"]),
                    feature_hint=FeatureHint(d["feature_hint"]) if d.get("feature_hint") else None,
                    feature_source=FeatureSource(d["feature_source"]) if d.get("feature_source") else None,
                    cardinality=d.get("cardinality"),
                )
            )
        return cls(features)
