"""Legacy id-indexing surface (reference experimental/preprocessing/
data_preparator.py: Spark ``Indexer``/``JoinBasedIndexer`` converting
arbitrary user/item ids to contiguous idx and back).  Pandas-native: wraps
two LabelEncodingRules; the Spark ML Estimator machinery has no MI355X
counterpart."""

from __future__ import annotations

import pandas as pd

from replay_amd.preprocessing import LabelEncodingRule


class Indexer:
    """Convert arbitrary ``user_id``/``item_id`` to contiguous
    ``user_idx``/``item_idx`` and back (reference data_preparator.py:33)."""

    def __init__(self, user_col: str = "user_id", item_col: str = "item_id") -> None:
        self.user_col = user_col
        self.item_col = item_col
        self.user_indexer = LabelEncodingRule(user_col, handle_unknown="use_default_value", default_value="last")
        self.item_indexer = LabelEncodingRule(item_col, handle_unknown="use_default_value", default_value="last")

    @property
    def _init_args(self):
        return {"user_col": self.user_col, "item_col": self.item_col}

    def fit(self, users: pd.DataFrame, items: pd.DataFrame) -> "Indexer":
        self.user_indexer.fit(users[[self.user_col]])
        self.item_indexer.fit(items[[self.item_col]])
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df
        if self.user_col in out.columns:
            out = self.user_indexer.transform(out).rename(columns={self.user_col: "user_idx"})
        if self.item_col in out.columns:
            out = self.item_indexer.transform(out).rename(columns={self.item_col: "item_idx"})
        return out

    def inverse_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df
        if "user_idx" in out.columns:
            out = out.rename(columns={"user_idx": self.user_col})
            out = self.user_indexer.inverse_transform(out)
        if "item_idx" in out.columns:
            out = out.rename(columns={"item_idx": self.item_col})
            out = self.item_indexer.inverse_transform(out)
        return out


class DataPreparator:
    """Legacy frame normalizer (reference data_preparator.py DataPreparator):
    renames mapped columns and casts timestamps, producing the canonical
    ``[user_id, item_id, timestamp, relevance]`` layout."""

    def __init__(self, columns_mapping: dict) -> None:
        self.columns_mapping = dict(columns_mapping)

    def transform(self, data: pd.DataFrame) -> pd.DataFrame:
        out = data.rename(columns={v: k for k, v in self.columns_mapping.items()})
        if "timestamp" in out.columns:
            if not pd.api.types.is_numeric_dtype(out["timestamp"]):
                out["timestamp"] = pd.to_datetime(out["timestamp"])
        if "relevance" not in out.columns:
            out["relevance"] = 1.0
        return out
