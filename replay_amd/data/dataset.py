"""Backend-polymorphic dataset container.

MI355X-native counterpart of the reference Dataset (replay/data/dataset.py:33)
with pandas (and optional polars) backends: consistency checks, unique-id
extraction, lazy cardinality, parquet save/load with schema JSON
(reference dataset.py:209,260,306), ``subset`` (dataset.py:397) and backend
conversion helpers (dataset.py:705-735).  The Spark backend intentionally
raises: the MI355X build's scale axis is GPUs, not JVM executors.
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, Iterable, Optional, Sequence, Union

import numpy as np
import pandas as pd

from replay_amd.utils.types import POLARS_AVAILABLE, DataFrameLike

from .schema import FeatureHint, FeatureInfo, FeatureSchema, FeatureSource, FeatureType

if POLARS_AVAILABLE:  # pragma: no cover
    import polars as pl


def _is_polars(df) -> bool:
    return POLARS_AVAILABLE and isinstance(df, pl.DataFrame)  # pragma: no cover


def _to_pandas(df) -> pd.DataFrame:
    if isinstance(df, pd.DataFrame):
        return df
    if _is_polars(df):  # pragma: no cover
        import warnings

        warnings.warn(
            "polars input is converted to pandas at the Dataset boundary: the "
            "MI355X build's single tabular execution backend is pandas/Arrow "
            "(a native polars pipeline is a documented descope — see "
            "docs/pages/parity.md). All downstream splitters/filters/metrics "
            "run the pandas path, which is oracle-equality-tested against the "
            "reference.",
            stacklevel=3,
        )
        return df.to_pandas()
    raise TypeError(f"Unsupported dataframe type: {type(df)} (Spark is not supported by the MI355X build)")


class Dataset:
    """Universal dataset: interactions plus optional query/item features."""

    def __init__(
        self,
        feature_schema: FeatureSchema,
        interactions: DataFrameLike,
        query_features: Optional[DataFrameLike] = None,
        item_features: Optional[DataFrameLike] = None,
        check_consistency: bool = True,
        categorical_encoded: bool = False,
    ) -> None:
        self._interactions = interactions
        self._query_features = query_features
        self._item_features = item_features
        self._categorical_encoded = categorical_encoded

        self._assign_df_type()
        self._feature_schema = self._fill_feature_schema(feature_schema)

        if check_consistency:
            if query_features is not None:
                self._check_ids_consistency(hint=FeatureHint.QUERY_ID)
            if item_features is not None:
                self._check_ids_consistency(hint=FeatureHint.ITEM_ID)
            if categorical_encoded:
                self._check_encoded()

    # -- basic properties ------------------------------------------------------
    def _assign_df_type(self) -> None:
        frames = [f for f in (self._interactions, self._query_features, self._item_features) if f is not None]
        kinds = {("polars" if _is_polars(f) else "pandas") for f in frames}
        if len(kinds) > 1:
            raise TypeError("All dataframes must share one backend")
        self.is_pandas = "pandas" in kinds
        self.is_polars = "polars" in kinds
        self.is_spark = False

    @property
    def interactions(self) -> DataFrameLike:
        return self._interactions

    @property
    def query_features(self) -> Optional[DataFrameLike]:
        return self._query_features

    @property
    def item_features(self) -> Optional[DataFrameLike]:
        return self._item_features

    @property
    def feature_schema(self) -> FeatureSchema:
        return self._feature_schema

    @property
    def is_categorical_encoded(self) -> bool:
        return self._categorical_encoded

    def __len__(self) -> int:
        return len(self._interactions)

    # -- id helpers ------------------------------------------------------------
    def _unique_of(self, column: str, extra_frame) -> np.ndarray:
        inter = _to_pandas(self._interactions)
        values = [inter[column].to_numpy()] if column in inter.columns else []
        if extra_frame is not None:
            extra = _to_pandas(extra_frame)
            if column in extra.columns:
                values.append(extra[column].to_numpy())
        if not values:
            return np.array([])
        return np.unique(np.concatenate(values))

    @property
    def query_ids(self) -> pd.DataFrame:
        col = self._feature_schema.query_id_column
        return pd.DataFrame({col: self._unique_of(col, self._query_features)})

    @property
    def item_ids(self) -> pd.DataFrame:
        col = self._feature_schema.item_id_column
        return pd.DataFrame({col: self._unique_of(col, self._item_features)})

    @property
    def query_count(self) -> int:
        return int(self._feature_schema.query_id_feature.cardinality)

    @property
    def item_count(self) -> int:
        return int(self._feature_schema.item_id_feature.cardinality)

    # -- schema wiring ---------------------------------------------------------
    def _source_of(self, column: str) -> Optional[FeatureSource]:
        inter_cols = set(_to_pandas(self._interactions).columns)
        if column in inter_cols:
            return FeatureSource.INTERACTIONS
        if self._query_features is not None and column in set(_to_pandas(self._query_features).columns):
            return FeatureSource.QUERY_FEATURES
        if self._item_features is not None and column in set(_to_pandas(self._item_features).columns):
            return FeatureSource.ITEM_FEATURES
        return None

    def _cardinality_impl(self, column: str) -> int:
        feature = self._feature_schema[column]
        if feature.feature_hint == FeatureHint.QUERY_ID:
            values = self._unique_of(column, self._query_features)
        elif feature.feature_hint == FeatureHint.ITEM_ID:
            values = self._unique_of(column, self._item_features)
        else:
            source = feature.feature_source
            frame = {
                FeatureSource.INTERACTIONS: self._interactions,
                FeatureSource.QUERY_FEATURES: self._query_features,
                FeatureSource.ITEM_FEATURES: self._item_features,
            }.get(source)
            frame = _to_pandas(frame)
            if feature.feature_type == FeatureType.CATEGORICAL_LIST:
                values = np.unique(np.concatenate([np.asarray(v) for v in frame[column]]))
            else:
                values = frame[column].unique()
        if self._categorical_encoded and len(values):
            # encoded ids are assumed contiguous [0, max]
            return int(np.max(values)) + 1
        return len(values)

    def _fill_feature_schema(self, schema: FeatureSchema) -> FeatureSchema:
        filled = schema.copy()
        for feature in filled:
            if feature.feature_source is None:
                source = self._source_of(feature.column)
                if source is None:
                    raise ValueError(f"Feature column {feature.column!r} not found in any dataframe")
                feature._set_feature_source(source)
            else:
                if self._source_of(feature.column) is None:
                    raise ValueError(f"Feature column {feature.column!r} not found in any dataframe")
            if feature.feature_type in (FeatureType.CATEGORICAL, FeatureType.CATEGORICAL_LIST):
                feature._set_cardinality_callback(self._cardinality_impl)
        return filled

    # -- checks ----------------------------------------------------------------
    def _check_ids_consistency(self, hint: FeatureHint) -> None:
        column = (
            self._feature_schema.query_id_column if hint == FeatureHint.QUERY_ID else self._feature_schema.item_id_column
        )
        features = self._query_features if hint == FeatureHint.QUERY_ID else self._item_features
        inter_ids = set(_to_pandas(self._interactions)[column].tolist())
        feat_ids = set(_to_pandas(features)[column].tolist())
        if inter_ids - feat_ids:
            raise ValueError(f"Interactions contain {column} values absent from the features frame")

    def _check_encoded(self) -> None:
        for feature in self._feature_schema.categorical_features:
            source = feature.feature_source
            frame = {
                FeatureSource.INTERACTIONS: self._interactions,
                FeatureSource.QUERY_FEATURES: self._query_features,
                FeatureSource.ITEM_FEATURES: self._item_features,
            }.get(source)
            if frame is None:
                continue
            col = _to_pandas(frame)[feature.column]
            if not pd.api.types.is_integer_dtype(col):
                raise ValueError(f"Column {feature.column} is not integer-encoded")
            if len(col) and col.min() < 0:
                raise ValueError(f"Column {feature.column} contains negative encoded ids")

    # -- subsetting ------------------------------------------------------------
    def subset(self, features_to_keep: Iterable[str]) -> "Dataset":
        keep = set(features_to_keep)
        keep.add(self._feature_schema.query_id_column)
        keep.add(self._feature_schema.item_id_column)
        schema = self._feature_schema.subset(keep)

        def _cols(frame, source):
            if frame is None:
                return None
            cols = [f.column for f in schema if f.feature_source == source]
            if not cols:
                return None
            frame = _to_pandas(frame)
            id_cols = [c for c in (schema.query_id_column, schema.item_id_column) if c in frame.columns]
            use = list(dict.fromkeys(id_cols + cols))
            return frame[[c for c in use if c in frame.columns]]

        interactions = _to_pandas(self._interactions)[
            [c for c in interactions_columns(schema) if c in _to_pandas(self._interactions).columns]
        ]
        return Dataset(
            feature_schema=schema,
            interactions=interactions,
            query_features=_cols(self._query_features, FeatureSource.QUERY_FEATURES),
            item_features=_cols(self._item_features, FeatureSource.ITEM_FEATURES),
            check_consistency=False,
            categorical_encoded=self._categorical_encoded,
        )

    # -- persistence -----------------------------------------------------------
    def save(self, path: Union[str, Path]) -> None:
        """Persist the dataset as parquet frames + schema/metadata JSON.

        Matches the reference ``Dataset.save`` layout intent
        (replay/data/dataset.py:260): a directory with parquet payloads and a
        JSON description sufficient for :meth:`load`.
        """
        base = Path(path).with_suffix(".replay") if not str(path).endswith(".replay") else Path(path)
        base.mkdir(parents=True, exist_ok=True)
        _to_pandas(self._interactions).to_parquet(base / "interactions.parquet", index=False)
        if self._query_features is not None:
            _to_pandas(self._query_features).to_parquet(base / "query_features.parquet", index=False)
        if self._item_features is not None:
            _to_pandas(self._item_features).to_parquet(base / "item_features.parquet", index=False)
        meta = {
            "feature_schema": self._feature_schema.to_dict(),
            "categorical_encoded": self._categorical_encoded,
            "has_query_features": self._query_features is not None,
            "has_item_features": self._item_features is not None,
        }
        (base / "metadata.json").write_text(json.dumps(meta, indent=2))

    @classmethod
    def load(cls, path: Union[str, Path], dataframe_type: Optional[str] = None) -> "Dataset":
        base = Path(path).with_suffix(".replay") if not str(path).endswith(".replay") else Path(path)
        meta = json.loads((base / "metadata.json").read_text())
        if dataframe_type not in (None, "pandas"):
            raise ValueError(f"Unsupported dataframe_type: {dataframe_type}")
        interactions = pd.read_parquet(base / "interactions.parquet")
        query_features = pd.read_parquet(base / "query_features.parquet") if meta["has_query_features"] else None
        item_features = pd.read_parquet(base / "item_features.parquet") if meta["has_item_features"] else None
        return cls(
            feature_schema=FeatureSchema.from_dict(meta["feature_schema"]),
            interactions=interactions,
            query_features=query_features,
            item_features=item_features,
            check_consistency=False,
            categorical_encoded=meta["categorical_encoded"],
        )

    # -- backend conversion ----------------------------------------------------
    def to_pandas(self) -> "Dataset":
        return Dataset(
            feature_schema=self._feature_schema.copy(),
            interactions=_to_pandas(self._interactions),
            query_features=_to_pandas(self._query_features) if self._query_features is not None else None,
            item_features=_to_pandas(self._item_features) if self._item_features is not None else None,
            check_consistency=False,
            categorical_encoded=self._categorical_encoded,
        )

    def to_polars(self) -> "Dataset":  # pragma: no cover
        if not POLARS_AVAILABLE:
            raise RuntimeError("polars is not installed")
        return Dataset(
            feature_schema=self._feature_schema.copy(),
            interactions=pl.from_pandas(_to_pandas(self._interactions)),
            query_features=pl.from_pandas(_to_pandas(self._query_features)) if self._query_features is not None else None,
            item_features=pl.from_pandas(_to_pandas(self._item_features)) if self._item_features is not None else None,
            check_consistency=False,
            categorical_encoded=self._categorical_encoded,
        )

    def to_spark(self):
        raise RuntimeError("Spark is not supported by the MI355X build of replay_amd")


def interactions_columns(schema: FeatureSchema) -> Sequence[str]:
    cols = [schema.query_id_column, schema.item_id_column]
    for f in schema:
        if f.feature_source == FeatureSource.INTERACTIONS and f.column not in cols:
            cols.append(f.column)
    return [c for c in cols if c is not None]


def nunique_per_column(df: pd.DataFrame) -> Dict[str, int]:
    return {c: int(df[c].nunique()) for c in df.columns}
