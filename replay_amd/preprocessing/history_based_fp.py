"""History-based feature generation.

Parity with reference replay/preprocessing/history_based_fp.py
(LogStatFeaturesProcessor:39, ConditionalPopularityProcessor:284,
HistoryBasedFeaturesProcessor:381): log-statistics features (counts, mean
ratings, recency, abnormality) for queries and items, plus conditional
popularity over categorical columns.  Pandas-native (the reference is Spark).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd


class EmptyFeatureProcessor:
    """No-op feature processor (reference history_based_fp.py:22): keeps the
    two-stage pipeline shape when no feature engineering is wanted."""

    def fit(self, log, features=None) -> "EmptyFeatureProcessor":
        return self

    def transform(self, log):
        return log


class LogStatFeaturesProcessor:
    def __init__(
        self,
        query_column: str = "query_id",
        item_column: str = "item_id",
        rating_column: str = "rating",
        timestamp_column: str = "timestamp",
    ) -> None:
        self.query_column = query_column
        self.item_column = item_column
        self.rating_column = rating_column
        self.timestamp_column = timestamp_column
        self.query_log_features: Optional[pd.DataFrame] = None
        self.item_log_features: Optional[pd.DataFrame] = None

    def fit(self, log: pd.DataFrame) -> "LogStatFeaturesProcessor":
        has_rating = self.rating_column in log.columns
        has_ts = self.timestamp_column in log.columns
        max_ts = log[self.timestamp_column].max() if has_ts else None

        def stats(by: str, prefix: str) -> pd.DataFrame:
            agg = log.groupby(by).agg(
                **{
                    f"{prefix}_log_num_interact": (self.item_column, "size"),
                }
            )
            if has_rating:
                agg[f"{prefix}_mean_rating"] = log.groupby(by)[self.rating_column].mean()
                agg[f"{prefix}_std_rating"] = log.groupby(by)[self.rating_column].std().fillna(0.0)
            if has_ts:
                last = log.groupby(by)[self.timestamp_column].max()
                agg[f"{prefix}_log_recency"] = np.log1p(max_ts - last)
            return agg.reset_index()

        self.query_log_features = stats(self.query_column, "u")
        self.item_log_features = stats(self.item_column, "i")

        if has_rating:
            # abnormality: mean |user rating - item mean rating|
            item_mean = log.groupby(self.item_column)[self.rating_column].transform("mean")
            ab = (log[self.rating_column] - item_mean).abs()
            abnormality = ab.groupby(log[self.query_column]).mean().rename("u_abnormality").reset_index()
            self.query_log_features = self.query_log_features.merge(abnormality, on=self.query_column)
        return self

    def transform(self, log: pd.DataFrame) -> pd.DataFrame:
        out = log.merge(self.query_log_features, on=self.query_column, how="left")
        out = out.merge(self.item_log_features, on=self.item_column, how="left")
        return out


class ConditionalPopularityProcessor:
    """Per-(entity, categorical-value) popularity (reference :284)."""

    def __init__(
        self,
        cat_features_list: List[str],
        entity_column: str = "item_id",
        query_column: str = "query_id",
    ) -> None:
        self.cat_features_list = cat_features_list
        self.entity_column = entity_column
        self.query_column = query_column
        self.conditional_pop_dict = {}

    def fit(self, log: pd.DataFrame, features: pd.DataFrame) -> "ConditionalPopularityProcessor":
        joined = log.merge(features, on=self.entity_column if self.entity_column in features.columns else self.query_column)
        for cat in self.cat_features_list:
            counts = joined.groupby([self.entity_column, cat]).size().rename("cnt").reset_index()
            totals = counts.groupby(self.entity_column)["cnt"].transform("sum")
            counts[f"pop_by_{cat}"] = counts["cnt"] / totals
            self.conditional_pop_dict[cat] = counts[[self.entity_column, cat, f"pop_by_{cat}"]]
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df
        for cat, pop in self.conditional_pop_dict.items():
            if cat in out.columns:
                out = out.merge(pop, on=[self.entity_column, cat], how="left")
                out[f"pop_by_{cat}"] = out[f"pop_by_{cat}"].fillna(0.0)
        return out


class HistoryBasedFeaturesProcessor:
    """Combined processor (reference :381)."""

    def __init__(
        self,
        use_log_features: bool = True,
        use_conditional_popularity: bool = True,
        query_cat_features_list: Optional[List[str]] = None,
        item_cat_features_list: Optional[List[str]] = None,
        query_column: str = "query_id",
        item_column: str = "item_id",
    ) -> None:
        self.use_log_features = use_log_features
        self.use_conditional_popularity = use_conditional_popularity
        self.log_processor = LogStatFeaturesProcessor(query_column=query_column, item_column=item_column)
        self.query_cond = (
            ConditionalPopularityProcessor(query_cat_features_list, entity_column=item_column)
            if query_cat_features_list
            else None
        )
        self.item_cond = (
            ConditionalPopularityProcessor(item_cat_features_list, entity_column=query_column)
            if item_cat_features_list
            else None
        )
        self.fitted = False

    def fit(self, log: pd.DataFrame, query_features: Optional[pd.DataFrame] = None, item_features: Optional[pd.DataFrame] = None):
        if self.use_log_features:
            self.log_processor.fit(log)
        if self.use_conditional_popularity and self.query_cond is not None and query_features is not None:
            self.query_cond.fit(log, query_features)
        if self.use_conditional_popularity and self.item_cond is not None and item_features is not None:
            self.item_cond.fit(log, item_features)
        self.fitted = True
        return self

    def transform(self, log: pd.DataFrame) -> pd.DataFrame:
        if not self.fitted:
            raise RuntimeError("Processor is not fitted")
        out = log
        if self.use_log_features:
            out = self.log_processor.transform(out)
        if self.query_cond is not None:
            out = self.query_cond.transform(out)
        if self.item_cond is not None:
            out = self.item_cond.transform(out)
        return out
