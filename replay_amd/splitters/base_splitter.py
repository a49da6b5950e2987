"""Splitter base class.

Parity with reference replay/splitters/base_splitter.py: ``split()`` (:169) =
abstract ``_core_split`` (:161) + optional cold-user/item dropping (:101-159),
plus session-id handling (:34-35) and JSON save/load (:72,87).
"""

from __future__ import annotations

import json
from pathlib import Path
from typing import Optional, Tuple, Union

import pandas as pd

SplitterReturnType = Tuple[pd.DataFrame, pd.DataFrame]


class Splitter:
    _init_arg_names: Tuple[str, ...] = (
        "drop_cold_users",
        "drop_cold_items",
        "query_column",
        "item_column",
        "timestamp_column",
        "session_id_column",
        "session_id_processing_strategy",
    )

    def __init__(
        self,
        drop_cold_items: bool = False,
        drop_cold_users: bool = False,
        query_column: str = "query_id",
        item_column: str = "item_id",
        timestamp_column: str = "timestamp",
        session_id_column: Optional[str] = None,
        session_id_processing_strategy: str = "test",
    ) -> None:
        self.drop_cold_users = drop_cold_users
        self.drop_cold_items = drop_cold_items
        self.query_column = query_column
        self.item_column = item_column
        self.timestamp_column = timestamp_column
        self.session_id_column = session_id_column
        if session_id_processing_strategy not in ("train", "test"):
            raise ValueError("session_id_processing_strategy must be 'train' or 'test'")
        self.session_id_processing_strategy = session_id_processing_strategy

    # -- public API ------------------------------------------------------------
    def split(self, interactions: pd.DataFrame) -> SplitterReturnType:
        train, test = self._core_split(interactions)
        train, test = self._drop_cold_entities(train, test)
        return train, test

    def _core_split(self, interactions: pd.DataFrame) -> SplitterReturnType:  # pragma: no cover
        raise NotImplementedError

    # -- helpers ---------------------------------------------------------------
    def _drop_cold_entities(self, train: pd.DataFrame, test: pd.DataFrame) -> SplitterReturnType:
        if self.drop_cold_items and len(test):
            test = test[test[self.item_column].isin(set(train[self.item_column]))]
        if self.drop_cold_users and len(test):
            test = test[test[self.query_column].isin(set(train[self.query_column]))]
        return train, test

    def _recalculate_with_session_id_column(
        self, interactions: pd.DataFrame, is_test_mask: pd.Series
    ) -> pd.Series:
        """Keep whole sessions on one side of the split.

        strategy='test': a session with any test row goes entirely to test;
        strategy='train': it goes entirely to train
        (reference base_splitter.py:34-35 semantics).
        """
        if self.session_id_column is None:
            return is_test_mask
        group_cols = [self.query_column, self.session_id_column]
        if self.session_id_processing_strategy == "test":
            session_has_test = is_test_mask.groupby(
                [interactions[c] for c in group_cols]
            ).transform("any")
            return session_has_test
        session_all_test = is_test_mask.groupby([interactions[c] for c in group_cols]).transform("all")
        return session_all_test

    # -- persistence -----------------------------------------------------------
    def save(self, path: Union[str, Path]) -> None:
        base = Path(path)
        base.mkdir(parents=True, exist_ok=True)
        args = {name: getattr(self, name) for name in self._init_arg_names}
        state = {"_class_name": type(self).__name__, "init_args": args}
        (base / "init_args.json").write_text(json.dumps(state, default=str))

    @classmethod
    def load(cls, path: Union[str, Path]) -> "Splitter":
        from replay_amd import splitters as _splitters

        base = Path(path)
        state = json.loads((base / "init_args.json").read_text())
        klass = getattr(_splitters, state["_class_name"])
        return klass(**state["init_args"])
