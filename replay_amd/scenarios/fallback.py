"""Fallback scenario: main model + fallback for queries with too few recs.

Parity with reference Fallback (replay/scenarios/fallback.py:13) and the
``fallback`` merge helper (replay/utils/spark_utils.py:480): fallback ratings
are shifted below the main model's minimum so main recs always rank first.
"""

from __future__ import annotations

from typing import Optional

import pandas as pd

from replay_amd.models.base_rec import BaseRecommender
from replay_amd.models.pop_rec import PopRec


class Fallback(BaseRecommender):
    def __init__(self, main_model: BaseRecommender, fallback_model: Optional[BaseRecommender] = None) -> None:
        super().__init__()
        self.main_model = main_model
        self.fb_model = fallback_model if fallback_model is not None else PopRec()

    def __str__(self) -> str:
        return f"Fallback({self.main_model}, {self.fb_model})"

    @property
    def _init_args(self):
        return {}

    def fit(self, dataset) -> None:
        self._fit_wrap(dataset)

    def _fit(self, dataset) -> None:
        self.main_model._fit_wrap(dataset)
        self.fb_model._fit_wrap(dataset)

    @staticmethod
    def _merge_recs(main: pd.DataFrame, extra: pd.DataFrame, query_column: str, rating_column: str) -> pd.DataFrame:
        """Shift extra ratings below main's minimum, then anti-join and concat
        (reference spark_utils.fallback:480)."""
        if main.empty:
            return extra
        if extra.empty:
            return main
        shift = main[rating_column].min() - extra[rating_column].max() - 1.0
        extra = extra.copy()
        extra[rating_column] = extra[rating_column] + shift
        item_column = [c for c in main.columns if c not in (query_column, rating_column)][0]
        merged = extra.merge(
            main[[query_column, item_column]].assign(__main=True),
            on=[query_column, item_column],
            how="left",
        )
        extra_only = merged[merged["__main"].isna()].drop(columns="__main")
        return pd.concat([main, extra_only], ignore_index=True)

    def predict(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        main_recs = self.main_model.predict(dataset, k, queries, items, filter_seen_items)
        fb_recs = self.fb_model.predict(dataset, k, queries, items, filter_seen_items)
        query_col = self.main_model.query_column
        rating_col = self.main_model.rating_column
        merged = self._merge_recs(main_recs, fb_recs, query_col, rating_col)
        merged = self.main_model._get_top_k(merged, query_col, rating_col, k)
        if recs_file_path is not None:
            merged.to_parquet(recs_file_path, index=False)
            return None
        return merged

    def fit_predict(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        self.fit(dataset)
        return self.predict(dataset, k, queries, items, filter_seen_items, recs_file_path)

    def optimize(self, train_dataset, test_dataset, param_borders=None, criterion=None, k=10, budget=10, new_study=True):
        """Optimize both models (reference fallback.py optimize)."""
        main_params = self.main_model.optimize(
            train_dataset, test_dataset, (param_borders or {}).get("main"), criterion, k, budget, new_study
        )
        fb_params = None
        if self.fb_model._search_space:
            fb_params = self.fb_model.optimize(
                train_dataset, test_dataset, (param_borders or {}).get("fallback"), criterion, k, budget, new_study
            )
        return main_params, fb_params
