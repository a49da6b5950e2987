"""Concrete splitters.

Parity with reference replay/splitters/: RatioSplitter (ratio_splitter.py:13),
LastNSplitter (last_n_splitter.py:24), TimeSplitter (time_splitter.py:20),
RandomSplitter (random_splitter.py:6), NewUsersSplitter
(new_users_splitter.py:12), ColdUserRandomSplitter
(cold_user_random_splitter.py:17), RandomNextNSplitter
(random_next_n_splitter.py:20), TwoStageSplitter (two_stage_splitter.py:17),
KFolds (k_folds.py:16).
"""

from __future__ import annotations

from typing import Iterator, Optional, Tuple, Union

import numpy as np
import pandas as pd

from .base_splitter import Splitter, SplitterReturnType


class RatioSplitter(Splitter):
    """Per-user temporal split: last ``test_size`` fraction of each user's
    interactions go to test."""

    _init_arg_names = Splitter._init_arg_names + ("test_size", "divide_column", "min_interactions_per_group")

    def __init__(
        self,
        test_size: float = 0.2,
        divide_column: Optional[str] = None,
        min_interactions_per_group: Optional[int] = None,
        split_by_fractions: bool = True,
        **kwargs,
    ) -> None:
        super().__init__(**kwargs)
        if not 0 < test_size < 1:
            raise ValueError("test_size must be in (0, 1)")
        self.test_size = test_size
        self.divide_column = divide_column or self.query_column
        self.min_interactions_per_group = min_interactions_per_group
        self.split_by_fractions = split_by_fractions

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        df = interactions.sort_values([self.divide_column, self.timestamp_column], kind="stable")
        sizes = df.groupby(self.divide_column)[self.divide_column].transform("size")
        row_num = df.groupby(self.divide_column).cumcount() + 1  # 1-based in time order
        if self.split_by_fractions:
            # reference ratio_splitter.py:215: frac = round(row/count, 3),
            # test rows are frac > round(1 - ratio, 3)
            frac = (row_num / sizes).round(3)
            is_test = frac > round(1.0 - self.test_size, 3)
        else:
            # non-fraction mode (reference ratio_splitter.py:292-309):
            # n_train = count - int(count * ratio); when no
            # min_interactions_per_group is set, groups too small to yield a
            # test row naturally (0 < count*ratio < 1) give up one anyway
            n_train = sizes - (sizes * self.test_size).astype(int)
            if self.min_interactions_per_group is None:
                frac = sizes * self.test_size
                n_train = n_train.where(~((frac > 0) & (frac < 1) & (n_train > 1)), n_train - 1)
            is_test = row_num > n_train
        if self.min_interactions_per_group is not None:
            # undersized groups go entirely to train (reference :214)
            is_test &= sizes >= self.min_interactions_per_group
        is_test = self._recalculate_with_session_id_column(df, is_test)
        return df[~is_test].sort_index(), df[is_test].sort_index()


class LastNSplitter(Splitter):
    """Last N interactions (or last N seconds) per user to test
    (reference last_n_splitter.py:24)."""

    _init_arg_names = Splitter._init_arg_names + ("N", "divide_column", "strategy")

    def __init__(
        self,
        N: int = 1,
        divide_column: Optional[str] = None,
        strategy: str = "interactions",
        **kwargs,
    ) -> None:
        super().__init__(**kwargs)
        if strategy not in ("interactions", "timedelta"):
            raise ValueError("strategy must be 'interactions' or 'timedelta'")
        self.N = N
        self.divide_column = divide_column or self.query_column
        self.strategy = strategy

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        df = interactions.sort_values([self.divide_column, self.timestamp_column], kind="stable")
        if self.strategy == "interactions":
            sizes = df.groupby(self.divide_column)[self.divide_column].transform("size")
            pos = df.groupby(self.divide_column).cumcount()
            is_test = pos >= (sizes - self.N)
        else:
            ts = df[self.timestamp_column]
            if pd.api.types.is_datetime64_any_dtype(ts):
                seconds = ts.astype("int64") // 10**9
            else:
                seconds = ts.astype("int64")
            last = seconds.groupby(df[self.divide_column]).transform("max")
            is_test = seconds > (last - self.N)
        is_test = self._recalculate_with_session_id_column(df, is_test)
        return df[~is_test].sort_index(), df[is_test].sort_index()


class TimeSplitter(Splitter):
    """Global time split at ``time_threshold`` (fraction or timestamp)
    (reference time_splitter.py:20)."""

    _init_arg_names = Splitter._init_arg_names + ("time_threshold",)

    def __init__(self, time_threshold, time_column_format: str = "%Y-%m-%d %H:%M:%S", **kwargs) -> None:
        super().__init__(**kwargs)
        self.time_threshold = time_threshold
        self.time_column_format = time_column_format

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        ts = interactions[self.timestamp_column]
        threshold = self.time_threshold
        if isinstance(threshold, float) and 0 < threshold < 1:
            sorted_ts = ts.sort_values()
            threshold = sorted_ts.iloc[min(len(sorted_ts) - 1, int(len(sorted_ts) * (1 - threshold)))]
            is_test = ts >= threshold
        else:
            if isinstance(threshold, str):
                threshold = pd.to_datetime(threshold, format=self.time_column_format)
                ts = pd.to_datetime(ts)
            is_test = ts >= threshold
        is_test = self._recalculate_with_session_id_column(interactions, is_test)
        return interactions[~is_test], interactions[is_test]


class RandomSplitter(Splitter):
    """Uniform random row split (reference random_splitter.py:6)."""

    _init_arg_names = Splitter._init_arg_names + ("test_size", "seed")

    def __init__(self, test_size: float = 0.2, seed: Optional[int] = None, **kwargs) -> None:
        super().__init__(**kwargs)
        if not 0 <= test_size <= 1:
            raise ValueError("test_size must be in [0, 1]")
        self.test_size = test_size
        self.seed = seed

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        # the reference's exact sampling call (random_splitter.py:56):
        # pandas .sample with the seed, so the split AND row order match
        train = interactions.sample(frac=(1 - self.test_size), random_state=self.seed)
        test = interactions.drop(train.index)
        if self.session_id_column:
            is_test = pd.Series(False, index=interactions.index)
            is_test[test.index] = True
            is_test = self._recalculate_with_session_id_column(interactions, is_test)
            return interactions[~is_test], interactions[is_test]
        return train, test


class NewUsersSplitter(Splitter):
    """Users whose first interaction is in the last ``test_size`` share of the
    timeline go entirely to test (reference new_users_splitter.py:12)."""

    _init_arg_names = Splitter._init_arg_names + ("test_size",)

    def __init__(self, test_size: float = 0.2, **kwargs) -> None:
        super().__init__(**kwargs)
        if not 0 < test_size < 1:
            raise ValueError("test_size must be in (0, 1)")
        self.test_size = test_size

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        # reference new_users_splitter.py:99: test_start is the LATEST first-
        # interaction date at which the cumulative (newest-first) user count
        # reaches test_size of all users; users starting at/after it go to
        # test WITH THEIR WHOLE HISTORY, and train keeps only interactions
        # strictly BEFORE test_start (old users' later rows are dropped)
        first_ts = interactions.groupby(self.query_column)[self.timestamp_column].min()
        by_date = (
            first_ts.value_counts().rename("n").sort_index(ascending=False).to_frame()
        )
        by_date["cum"] = by_date["n"].cumsum()
        eligible = by_date[by_date["cum"] >= self.test_size * by_date["n"].sum()]
        test_start = eligible.index.max()
        new_users = set(first_ts[first_ts >= test_start].index)
        is_test = interactions[self.query_column].isin(new_users)
        in_train = interactions[self.timestamp_column] < test_start
        is_test = self._recalculate_with_session_id_column(interactions, is_test)
        return interactions[~is_test & in_train], interactions[is_test]


class ColdUserRandomSplitter(Splitter):
    """A random ``test_size`` fraction of users moves entirely to test
    (reference cold_user_random_splitter.py:17)."""

    _init_arg_names = Splitter._init_arg_names + ("test_size", "seed")

    def __init__(self, test_size: float = 0.2, seed: Optional[int] = None, **kwargs) -> None:
        super().__init__(**kwargs)
        self.test_size = test_size
        self.seed = seed

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        # the reference's exact sampling (cold_user_random_splitter.py:64):
        # a pandas .sample over the unique-user frame with the seed
        users = pd.DataFrame(interactions[self.query_column].unique(), columns=[self.query_column])
        train_users = set(users.sample(frac=(1 - self.test_size), random_state=self.seed)[self.query_column])
        is_test = ~interactions[self.query_column].isin(train_users)
        return interactions[~is_test], interactions[is_test]


class RandomNextNSplitter(Splitter):
    """Pick a random cut position per user; the following N interactions form
    the test part (reference random_next_n_splitter.py:20)."""

    _init_arg_names = Splitter._init_arg_names + ("N", "seed", "divide_column")

    def __init__(self, N: int = 1, seed: Optional[int] = None, divide_column: Optional[str] = None, **kwargs) -> None:
        super().__init__(**kwargs)
        self.N = N
        self.seed = seed
        self.divide_column = divide_column or self.query_column

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        """Exact reference algorithm (random_next_n_splitter.py:121-146):
        cut = RandomState(seed).randint(0, count) per divide group (group
        order = first occurrence in the time-sorted frame; a cut of 0 sends
        the whole sequence to test), rows past cut+N are dropped, and rows
        at rank >= cut go to test."""
        df = interactions.sort_values([self.divide_column, self.timestamp_column])
        rank = df.groupby(self.divide_column, sort=False).cumcount()
        counts = df.groupby(self.divide_column, sort=False).size()
        rng = np.random.RandomState(self.seed)
        cuts = pd.Series(rng.randint(0, counts.values), index=counts.index)
        cut = df[self.divide_column].map(cuts)
        if self.N is not None:
            keep = rank < cut + self.N
            df, rank, cut = df[keep], rank[keep], cut[keep]
        is_test = rank >= cut
        is_test = self._recalculate_with_session_id_column(df, is_test)
        return df[~is_test][interactions.columns], df[is_test][interactions.columns]


class TwoStageSplitter(Splitter):
    """Two-stage split (reference two_stage_splitter.py:17, exact pandas
    algorithm): ``first_divide_size`` selects the TEST USERS (count if >= 1,
    fraction otherwise; seeded pandas .sample over the sorted unique users);
    ``second_divide_size`` then takes each test user's newest interactions —
    a per-user fraction when in [0, 1), or a fixed count when an int >= 1.
    ``shuffle=True`` ranks randomly instead of by recency."""

    _init_arg_names = Splitter._init_arg_names + (
        "first_divide_size", "second_divide_size", "first_divide_column", "shuffle", "seed",
    )

    def __init__(
        self,
        first_divide_size: Union[float, int] = 0.5,
        second_divide_size: Union[float, int] = 0.5,
        first_divide_column: str = "query_id",
        shuffle: bool = False,
        seed: Optional[int] = None,
        **kwargs,
    ) -> None:
        super().__init__(**kwargs)
        self.first_divide_size = first_divide_size
        self.second_divide_size = second_divide_size
        self.first_divide_column = first_divide_column
        self.shuffle = shuffle
        self.seed = seed

    def _get_test_values(self, interactions: pd.DataFrame) -> pd.DataFrame:
        all_values = pd.DataFrame(
            np.sort(interactions[self.first_divide_column].unique()),
            columns=[self.first_divide_column],
        )
        user_count = len(all_values)
        if isinstance(self.first_divide_size, int) and not isinstance(self.first_divide_size, bool):
            if not 1 <= self.first_divide_size < user_count:
                raise ValueError(f"Invalid value for user_test_size: {self.first_divide_size}")
            test_user_count = self.first_divide_size
        else:
            if not 0 < self.first_divide_size < 1:
                raise ValueError(f"Invalid value for user_test_size: {self.first_divide_size}")
            test_user_count = user_count * self.first_divide_size
        return all_values.sample(n=int(test_user_count), random_state=self.seed)

    def _partition(self, merged: pd.DataFrame) -> pd.DataFrame:
        if self.shuffle:
            res = merged.sample(frac=1, random_state=self.seed).sort_values(self.first_divide_column)
            res["_row_num"] = res.groupby(self.first_divide_column, sort=False).cumcount() + 1
            return res
        res = merged.copy(deep=True)
        res.sort_values([self.query_column, self.timestamp_column], ascending=[True, False], inplace=True)
        res["_row_num"] = res.groupby(self.query_column, sort=False).cumcount() + 1
        return res

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        test_users = self._get_test_values(interactions)
        test_users = test_users.assign(is_test=True)
        res = self._partition(interactions.merge(test_users, how="left", on=self.first_divide_column))
        res["is_test"] = res["is_test"].notna() & res["is_test"].eq(True)
        drop_cols = ["_row_num", "is_test"]
        if 0 <= self.second_divide_size < 1.0:
            counts = res.groupby(self.first_divide_column)[self.first_divide_column].transform("size")
            frac = res["_row_num"] / counts
            is_test = (frac <= self.second_divide_size) & res["is_test"]
            keep_train = (frac > self.second_divide_size) | (~res["is_test"])
        elif self.second_divide_size >= 1 and isinstance(self.second_divide_size, int):
            is_test = (res["_row_num"] <= self.second_divide_size) & res["is_test"]
            keep_train = (res["_row_num"] > self.second_divide_size) | (~res["is_test"])
        else:
            raise ValueError(
                f"`test_size` value must be [0, 1) or a positive integer; "
                f"test_size={self.second_divide_size}"
            )
        train = res[keep_train].drop(columns=drop_cols)
        test = res[is_test].drop(columns=drop_cols)
        return train, test


class KFolds:
    """K-fold user-wise random splitter (reference k_folds.py:16).

    Yields (train, test) pairs; fold assignment is per interaction within a
    user (strategy='query').
    """

    def __init__(
        self,
        n_folds: int = 5,
        strategy: str = "query",
        seed: Optional[int] = None,
        query_column: str = "query_id",
        item_column: str = "item_id",
        timestamp_column: str = "timestamp",
        session_id_column: Optional[str] = None,
        session_id_processing_strategy: str = "test",
    ) -> None:
        if strategy not in ("query",):
            raise ValueError("strategy must be 'query'")
        self.n_folds = n_folds
        self.strategy = strategy
        self.seed = seed
        self.query_column = query_column

    def split(self, interactions: pd.DataFrame) -> Iterator[Tuple[pd.DataFrame, pd.DataFrame]]:
        # the reference's exact fold assignment (k_folds.py:104): global
        # seeded shuffle, per-user cumcount + 1 modulo n_folds
        df = interactions.sample(frac=1, random_state=self.seed).sort_values(self.query_column)
        fold = (df.groupby(self.query_column, sort=False).cumcount() + 1) % self.n_folds
        for k in range(self.n_folds):
            mask = fold == k
            yield df[~mask], df[mask]
