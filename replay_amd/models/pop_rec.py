"""Popularity / random / category-popularity baselines.

Parity: PopRec (reference replay/models/pop_rec.py:10), QueryPopRec
(query_pop_rec.py), RandomRec (random_rec.py:10), CatPopRec (cat_pop_rec.py).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from .base_rec import NonPersonalizedRecommender, QueryRecommender, Recommender


class PopRec(NonPersonalizedRecommender):
    """Item popularity = share of users who interacted with the item
    (or rating sum/mean with use_rating)."""

    def __init__(self, use_rating: bool = False, add_cold_items: bool = True, cold_weight: float = 0.5) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        self.use_rating = use_rating

    @property
    def _init_args(self):
        return {
            "use_rating": self.use_rating,
            "add_cold_items": self.add_cold_items,
            "cold_weight": self.cold_weight,
        }

    _search_space = {"use_rating": {"type": "categorical", "args": [True, False]}}

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        n_queries = max(1, self._num_queries)
        if self.use_rating and self.rating_column in inter.columns:
            pop = inter.groupby(self.item_column)[self.rating_column].sum() / n_queries
        else:
            pop = inter.groupby(self.item_column)[self.query_column].nunique() / n_queries
        self.item_popularity = pop.rename(self.rating_column).reset_index()


class QueryPopRec(QueryRecommender):
    """Recommends each query its own most frequent items
    (reference query_pop_rec.py).  Intended for repeat-consumption data."""

    def __init__(self) -> None:
        super().__init__()
        self.query_item_popularity: Optional[pd.DataFrame] = None

    @property
    def _init_args(self):
        return {}

    @property
    def _dataframes(self):
        return {"query_item_popularity": self.query_item_popularity}

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        counts = inter.groupby([self.query_column, self.item_column]).size().rename("count").reset_index()
        totals = counts.groupby(self.query_column)["count"].transform("sum")
        counts[self.rating_column] = counts["count"] / totals
        self.query_item_popularity = counts[[self.query_column, self.item_column, self.rating_column]]

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        recs = self.query_item_popularity.merge(queries, on=self.query_column).merge(
            items, on=self.item_column
        )
        return recs


class RandomRec(NonPersonalizedRecommender):
    """Random recommendations; distribution in {uniform, popular_based,
    relevance} (reference random_rec.py:10)."""

    def __init__(
        self,
        distribution: str = "uniform",
        alpha: float = 0.0,
        seed: Optional[int] = None,
        add_cold_items: bool = True,
        cold_weight: float = 0.5,
    ) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        if distribution not in ("uniform", "popular_based", "relevance"):
            raise ValueError("distribution must be uniform/popular_based/relevance")
        if alpha < -1.0 and distribution == "popular_based":
            raise ValueError("alpha must be >= -1")
        self.distribution = distribution
        self.alpha = alpha
        self.seed = seed

    @property
    def _init_args(self):
        return {
            "distribution": self.distribution,
            "alpha": self.alpha,
            "seed": self.seed,
            "add_cold_items": self.add_cold_items,
            "cold_weight": self.cold_weight,
        }

    _search_space = {
        "distribution": {"type": "categorical", "args": ["uniform", "popular_based"]},
        "alpha": {"type": "uniform", "args": [-0.5, 100]},
    }

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        if self.distribution == "popular_based":
            pop = inter.groupby(self.item_column)[self.query_column].nunique().astype(float) + self.alpha
            pop = pop.clip(lower=1e-9)
        elif self.distribution == "relevance" and self.rating_column in inter.columns:
            pop = inter.groupby(self.item_column)[self.rating_column].sum().astype(float).clip(lower=1e-9)
        else:
            pop = pd.Series(1.0, index=pd.Index(inter[self.item_column].unique(), name=self.item_column))
        pop = pop / pop.sum()
        self.item_popularity = pop.rename(self.rating_column).reset_index()

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        rng = np.random.default_rng(self.seed)
        scored_items = self._fill_cold_ratings(items)
        probs = scored_items[self.rating_column].to_numpy(dtype=np.float64)
        probs = probs / probs.sum() if probs.sum() > 0 else np.full(len(probs), 1 / max(1, len(probs)))
        item_ids = scored_items[self.item_column].to_numpy()
        n = min(k, len(item_ids))
        rows = []
        for q in queries[self.query_column].to_numpy():
            chosen = rng.choice(len(item_ids), size=n, replace=False, p=probs)
            ratings = 1.0 / np.arange(1, n + 1)
            rows.append(
                pd.DataFrame(
                    {self.query_column: q, self.item_column: item_ids[chosen], self.rating_column: ratings}
                )
            )
        return pd.concat(rows, ignore_index=True) if rows else pd.DataFrame(
            columns=[self.query_column, self.item_column, self.rating_column]
        )


class CatPopRec(Recommender):
    """Per-category popularity (reference cat_pop_rec.py): recommends the most
    popular items inside each requested category."""

    can_predict_cold_queries = True

    def __init__(self, cat_tree: Optional[pd.DataFrame] = None, category_column: str = "category") -> None:
        super().__init__()
        self.category_column = category_column
        self.cat_tree = cat_tree
        self.cat_item_popularity: Optional[pd.DataFrame] = None

    @property
    def _init_args(self):
        return {"category_column": self.category_column}

    @property
    def _dataframes(self):
        return {"cat_item_popularity": self.cat_item_popularity}

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        if self.category_column not in inter.columns:
            raise ValueError(f"Interactions must contain the {self.category_column!r} column")
        counts = (
            inter.groupby([self.category_column, self.item_column]).size().rename("count").reset_index()
        )
        totals = counts.groupby(self.category_column)["count"].transform("sum")
        counts[self.rating_column] = counts["count"] / totals
        self.cat_item_popularity = counts[[self.category_column, self.item_column, self.rating_column]]

    def predict(self, categories: pd.DataFrame, k: int) -> pd.DataFrame:
        """Top-k items per requested category."""
        merged = self.cat_item_popularity.merge(categories, on=self.category_column)
        merged = merged.sort_values(
            [self.category_column, self.rating_column], ascending=[True, False], kind="stable"
        )
        return merged.groupby(self.category_column, sort=False).head(k).reset_index(drop=True)

    def _predict(self, dataset, k, queries, items, filter_seen_items=True):  # pragma: no cover
        raise NotImplementedError("CatPopRec predicts per category; use predict(categories, k)")
