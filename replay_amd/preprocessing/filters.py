"""Interaction-log filters.

Parity set with reference replay/preprocessing/filters.py:57-996:
InteractionEntriesFilter, MinCountFilter, LowRatingFilter,
NumInteractionsFilter, EntityDaysFilter, GlobalDaysFilter, TimePeriodFilter,
QuantileItemsFilter, ConsecutiveDuplicatesFilter.  Pandas/numpy-native.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd


class _BaseFilter:
    """Common transform() entry point."""

    def transform(self, interactions: pd.DataFrame) -> pd.DataFrame:
        return self._core_filter(interactions.copy())

    def _core_filter(self, interactions: pd.DataFrame) -> pd.DataFrame:  # pragma: no cover
        raise NotImplementedError


class InteractionEntriesFilter(_BaseFilter):
    """Iteratively remove interactions until every user and item falls inside
    the given [min, max] interaction-count bounds
    (reference filters.py:57)."""

    def __init__(
        self,
        query_column: str = "user_id",
        item_column: str = "item_id",
        min_inter_per_user: Optional[int] = None,
        max_inter_per_user: Optional[int] = None,
        min_inter_per_item: Optional[int] = None,
        max_inter_per_item: Optional[int] = None,
        allow_caching: bool = True,
    ) -> None:
        self.query_column = query_column
        self.item_column = item_column
        self.min_inter_per_user = min_inter_per_user
        self.max_inter_per_user = max_inter_per_user
        self.min_inter_per_item = min_inter_per_item
        self.max_inter_per_item = max_inter_per_item

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        changed = True
        while changed and len(df):
            changed = False
            user_counts = df.groupby(self.query_column)[self.item_column].transform("size")
            mask = pd.Series(True, index=df.index)
            if self.min_inter_per_user is not None:
                mask &= user_counts >= self.min_inter_per_user
            if self.max_inter_per_user is not None:
                mask &= user_counts <= self.max_inter_per_user
            if not mask.all():
                df = df[mask]
                changed = True
            if not len(df):
                break
            item_counts = df.groupby(self.item_column)[self.query_column].transform("size")
            mask = pd.Series(True, index=df.index)
            if self.min_inter_per_item is not None:
                mask &= item_counts >= self.min_inter_per_item
            if self.max_inter_per_item is not None:
                mask &= item_counts <= self.max_inter_per_item
            if not mask.all():
                df = df[mask]
                changed = True
        return df


class MinCountFilter(_BaseFilter):
    """Keep entities with at least ``num_entries`` interactions
    (reference filters.py:253)."""

    def __init__(self, num_entries: int, groupby_column: str = "user_id") -> None:
        self.num_entries = num_entries
        self.groupby_column = groupby_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        counts = df.groupby(self.groupby_column)[self.groupby_column].transform("size")
        return df[counts >= self.num_entries]


class LowRatingFilter(_BaseFilter):
    """Keep interactions with rating >= value (reference filters.py:315)."""

    def __init__(self, value: float, rating_column: str = "rating") -> None:
        self.value = value
        self.rating_column = rating_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        return df[df[self.rating_column] >= self.value]


class NumInteractionsFilter(_BaseFilter):
    """Keep the first/last ``num_interactions`` per user sorted by timestamp
    (reference filters.py:352)."""

    def __init__(
        self,
        num_interactions: int = 10,
        first: bool = True,
        query_column: str = "user_id",
        timestamp_column: str = "timestamp",
        item_column: Optional[str] = "item_id",
    ) -> None:
        self.num_interactions = num_interactions
        self.first = first
        self.query_column = query_column
        self.timestamp_column = timestamp_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        df = df.sort_values([self.query_column, self.timestamp_column], ascending=[True, self.first])
        picked = df.groupby(self.query_column).head(self.num_interactions)
        return picked.sort_index()


class EntityDaysFilter(_BaseFilter):
    """Keep the first/last ``days`` of interactions per entity
    (reference filters.py:494)."""

    def __init__(
        self,
        days: int = 10,
        first: bool = True,
        entity_column: str = "user_id",
        timestamp_column: str = "timestamp",
    ) -> None:
        self.days = days
        self.first = first
        self.entity_column = entity_column
        self.timestamp_column = timestamp_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        ts = pd.to_datetime(df[self.timestamp_column])
        delta = pd.Timedelta(days=self.days)
        if self.first:
            bound = ts.groupby(df[self.entity_column]).transform("min") + delta
            return df[ts < bound]
        bound = ts.groupby(df[self.entity_column]).transform("max") - delta
        return df[ts > bound]


class GlobalDaysFilter(_BaseFilter):
    """Keep the first/last ``days`` of the whole log (reference filters.py:633)."""

    def __init__(self, days: int = 10, first: bool = True, timestamp_column: str = "timestamp") -> None:
        self.days = days
        self.first = first
        self.timestamp_column = timestamp_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        ts = pd.to_datetime(df[self.timestamp_column])
        delta = pd.Timedelta(days=self.days)
        if self.first:
            return df[ts < ts.min() + delta]
        return df[ts > ts.max() - delta]


class TimePeriodFilter(_BaseFilter):
    """Keep interactions within [start_date, end_date)
    (reference filters.py:735)."""

    def __init__(
        self,
        start_date=None,
        end_date=None,
        timestamp_column: str = "timestamp",
        time_column_format: str = "%Y-%m-%d %H:%M:%S",
    ) -> None:
        self.start_date = start_date
        self.end_date = end_date
        self.timestamp_column = timestamp_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        ts = df[self.timestamp_column]
        if pd.api.types.is_datetime64_any_dtype(ts) or isinstance(self.start_date, str):
            ts = pd.to_datetime(ts)
            start = pd.to_datetime(self.start_date) if self.start_date is not None else None
            end = pd.to_datetime(self.end_date) if self.end_date is not None else None
        else:
            start, end = self.start_date, self.end_date
        mask = pd.Series(True, index=df.index)
        if start is not None:
            mask &= ts >= start
        if end is not None:
            mask &= ts < end
        return df[mask]


class QuantileItemsFilter(_BaseFilter):
    """Down-sample interactions of over-popular items above the ``alpha_quantile``
    item-popularity quantile, keeping at most ``items_proportion`` of each such
    item's interactions (reference filters.py:833)."""

    def __init__(
        self,
        alpha_quantile: float = 0.99,
        items_proportion: float = 0.5,
        query_column: str = "user_id",
        item_column: str = "item_id",
    ) -> None:
        if not 0 < alpha_quantile < 1:
            raise ValueError("alpha_quantile must be in (0, 1)")
        if not 0 < items_proportion < 1:
            raise ValueError("items_proportion must be in (0, 1)")
        self.alpha_quantile = alpha_quantile
        self.items_proportion = items_proportion
        self.query_column = query_column
        self.item_column = item_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        """Exact reference algorithm (filters.py:900-922): the popularity
        threshold is the midpoint-interpolated quantile of item counts; for
        each over-threshold item, items_proportion * (count - max long-tail
        count) interactions are deleted, taken from the MOST ACTIVE users
        first (keeps the item-popularity ordering intact)."""
        item_counts = df.groupby(self.item_column)[self.item_column].transform("size")
        user_counts = df.groupby(self.query_column)[self.query_column].transform("size")
        per_item = df.groupby(self.item_column).size()
        threshold = per_item.quantile(self.alpha_quantile, interpolation="midpoint")
        long_tail_mask = item_counts <= threshold
        long_tail = df[long_tail_mask]
        short = df[~long_tail_mask].copy()
        if not len(short):
            return df
        long_tail_max = int(item_counts[long_tail_mask].max()) if long_tail_mask.any() else 0
        short["_n_del"] = (
            self.items_proportion * (item_counts[~long_tail_mask] - long_tail_max)
        ).astype(int)
        short["_u_cnt"] = user_counts[~long_tail_mask]
        short = short.sort_values("_u_cnt", ascending=False)  # reference tie order (quicksort)

        def keep_mask(x):
            mask = np.ones(len(x), dtype=bool)
            mask[: int(x.iloc[0])] = False
            return pd.Series(mask, index=x.index)

        mask = short.groupby(self.item_column)["_n_del"].transform(keep_mask).astype(bool)
        kept = short.loc[mask, df.columns.tolist()]
        return pd.concat([long_tail, kept])


class ConsecutiveDuplicatesFilter(_BaseFilter):
    """Remove immediate repeats of the same item inside each user's
    time-ordered sequence (reference filters.py:996)."""

    def __init__(
        self,
        keep: str = "first",
        query_column: str = "user_id",
        item_column: str = "item_id",
        timestamp_column: str = "timestamp",
    ) -> None:
        if keep not in ("first", "last"):
            raise ValueError("keep must be 'first' or 'last'")
        self.keep = keep
        self.query_column = query_column
        self.item_column = item_column
        self.timestamp_column = timestamp_column

    def _core_filter(self, df: pd.DataFrame) -> pd.DataFrame:
        df = df.sort_values([self.query_column, self.timestamp_column], kind="stable")
        same_user = df[self.query_column].eq(df[self.query_column].shift())
        same_item = df[self.item_column].eq(df[self.item_column].shift())
        if self.keep == "first":
            mask = ~(same_user & same_item)
        else:
            same_user_next = df[self.query_column].eq(df[self.query_column].shift(-1))
            same_item_next = df[self.item_column].eq(df[self.item_column].shift(-1))
            mask = ~(same_user_next & same_item_next)
        return df[mask]
