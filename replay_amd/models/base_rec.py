"""Recommender base-class tree.

Parity with the reference class tree (replay/models/base_rec.py): IsSavable:52,
BaseRecommender:86 (`_fit_wrap`:99, `_predict_wrap`:258, `_filter_seen`:152,
`_filter_cold_for_predict`:316, `_predict_pairs_wrap`:502),
ItemVectorModel:692, HybridRecommender:795, Recommender:926 (public fit:929 /
predict:939 / predict_pairs:976 / fit_predict:1004), QueryRecommender:1052,
NonPersonalizedRecommender:1143.

The MI355X build is pandas/numpy-native on CPU with optional torch/HIP scoring
for matrix-factor models; there is no Spark tier.  Recommendation frames are
pandas DataFrames [query_column, item_column, rating].
"""

from __future__ import annotations

import json
import logging
from os.path import join
from pathlib import Path
from typing import Any, Dict, Iterable, Optional, Sequence, Union

import numpy as np
import pandas as pd

logger = logging.getLogger("replay_amd")


class IsSavable:
    """Save/load contract (reference base_rec.py:52)."""

    @property
    def _init_args(self) -> Dict[str, Any]:  # pragma: no cover
        raise NotImplementedError

    @property
    def _dataframes(self) -> Dict[str, Optional[pd.DataFrame]]:
        return {}

    def _save_model(self, path: str) -> None:
        pass

    def _load_model(self, path: str) -> None:
        pass

    def save(self, path: Union[str, Path]) -> None:
        base = Path(path)
        base.mkdir(parents=True, exist_ok=True)
        meta = {"_class_name": type(self).__name__, "init_args": self._init_args}
        (base / "init_args.json").write_text(json.dumps(meta, default=str))
        df_dir = base / "dataframes"
        df_dir.mkdir(exist_ok=True)
        for name, df in self._dataframes.items():
            if df is not None:
                df.to_parquet(df_dir / f"{name}.parquet", index=False)
        fit_info = {}
        for attr in ("fit_queries", "fit_items"):
            val = getattr(self, attr, None)
            if val is not None:
                val.to_parquet(base / f"{attr}.parquet", index=False)
                fit_info[attr] = True
        for attr in ("_num_queries", "_num_items", "_query_dim_size", "_item_dim_size", "query_column", "item_column", "rating_column", "timestamp_column"):
            if hasattr(self, attr):
                fit_info[attr] = getattr(self, attr)
        (base / "fit_info.json").write_text(json.dumps(fit_info, default=str))
        self._save_model(str(base))

    def _restore(self, path: Union[str, Path]) -> None:
        base = Path(path)
        df_dir = base / "dataframes"
        if df_dir.exists():
            for f in df_dir.glob("*.parquet"):
                setattr(self, f.stem, pd.read_parquet(f))
        fit_info_path = base / "fit_info.json"
        if fit_info_path.exists():
            fit_info = json.loads(fit_info_path.read_text())
            for attr in ("fit_queries", "fit_items"):
                if fit_info.pop(attr, None):
                    setattr(self, attr, pd.read_parquet(base / f"{attr}.parquet"))
            for attr, val in fit_info.items():
                if attr.startswith("_num") or attr.endswith("_dim_size"):
                    val = int(val) if val is not None and val != "None" else None
                setattr(self, attr, val)
        self._load_model(str(base))


class BaseRecommender(IsSavable):
    """Common fit/predict plumbing (reference base_rec.py:86)."""

    can_predict_cold_queries: bool = False
    can_predict_cold_items: bool = False
    _search_space: Optional[Dict[str, Any]] = None

    def __init__(self) -> None:
        self.query_column: str = "query_id"
        self.item_column: str = "item_id"
        self.rating_column: str = "rating"
        self.timestamp_column: str = "timestamp"
        self.fit_queries: Optional[pd.DataFrame] = None
        self.fit_items: Optional[pd.DataFrame] = None
        self._num_queries: int = 0
        self._num_items: int = 0
        self._query_dim_size: int = 0
        self._item_dim_size: int = 0

    # -- properties ------------------------------------------------------------
    @property
    def _init_args(self) -> Dict[str, Any]:
        return {}

    @property
    def queries_count(self) -> int:
        return self._num_queries

    @property
    def items_count(self) -> int:
        return self._num_items

    @property
    def logger(self):
        return logger

    def __str__(self) -> str:
        return type(self).__name__

    # -- fitting ---------------------------------------------------------------
    def _fit_wrap(self, dataset) -> None:
        """Extract columns, compute fit_queries/fit_items and dims, then
        delegate to the model-specific ``_fit`` (reference base_rec.py:99)."""
        schema = dataset.feature_schema
        self.query_column = schema.query_id_column
        self.item_column = schema.item_id_column
        self.rating_column = schema.interactions_rating_column or "rating"
        self.timestamp_column = schema.interactions_timestamp_column or "timestamp"

        self.fit_queries = dataset.query_ids
        self.fit_items = dataset.item_ids
        self._num_queries = len(self.fit_queries)
        self._num_items = len(self.fit_items)
        self._query_dim_size = int(self.fit_queries[self.query_column].max()) + 1 if self._num_queries else 0
        self._item_dim_size = int(self.fit_items[self.item_column].max()) + 1 if self._num_items else 0
        self._fit(dataset)

    def _fit(self, dataset) -> None:  # pragma: no cover
        raise NotImplementedError

    # -- predict plumbing -------------------------------------------------------
    @staticmethod
    def _ids_frame(ids, column: str) -> pd.DataFrame:
        if ids is None:
            return None
        if isinstance(ids, pd.DataFrame):
            return ids[[column]].drop_duplicates()
        if hasattr(ids, "interactions"):
            return pd.DataFrame({column: ids.interactions[column].unique()})
        return pd.DataFrame({column: pd.unique(pd.Series(list(ids)))})

    def _filter_cold_for_predict(self, queries: pd.DataFrame, items: pd.DataFrame):
        """Drop queries/items unseen at fit time unless the model supports them
        (reference base_rec.py:316)."""
        if not self.can_predict_cold_queries:
            known = set(self.fit_queries[self.query_column])
            n_before = len(queries)
            queries = queries[queries[self.query_column].isin(known)]
            if len(queries) < n_before:
                logger.info("%s dropped %d cold queries", self, n_before - len(queries))
        if not self.can_predict_cold_items:
            known = set(self.fit_items[self.item_column])
            n_before = len(items)
            items = items[items[self.item_column].isin(known)]
            if len(items) < n_before:
                logger.info("%s dropped %d cold items", self, n_before - len(items))
        return queries, items

    def _filter_seen(self, recs: pd.DataFrame, interactions: pd.DataFrame, queries: pd.DataFrame) -> pd.DataFrame:
        """Anti-join recommendations against the query's seen items
        (reference base_rec.py:152-201).  Integer-id frames take a numpy
        combined-key isin (the pandas merge was the predict bottleneck for
        dense-similarity models at ML-1M scale)."""
        qc, ic = self.query_column, self.item_column
        seen = interactions[interactions[qc].isin(set(queries[qc]))][[qc, ic]]
        if (
            pd.api.types.is_integer_dtype(recs[qc])
            and pd.api.types.is_integer_dtype(recs[ic])
            and pd.api.types.is_integer_dtype(seen[qc])
            and pd.api.types.is_integer_dtype(seen[ic])
            and len(recs)
        ):
            scale = int(max(recs[ic].max(), seen[ic].max() if len(seen) else 0)) + 1
            rk = recs[qc].to_numpy(np.int64) * scale + recs[ic].to_numpy(np.int64)
            sk = seen[qc].to_numpy(np.int64) * scale + seen[ic].to_numpy(np.int64)
            return recs[~np.isin(rk, sk)]
        merged = recs.merge(seen.assign(__seen=True), on=[qc, ic], how="left")
        return merged[merged["__seen"].isna()].drop(columns="__seen")

    @staticmethod
    def _get_top_k(recs: pd.DataFrame, query_column: str, rating_column: str, k: int) -> pd.DataFrame:
        """Per-query top-k by rating: one stable numpy lexsort + rank-within-
        group instead of pandas sort_values + groupby-head (several-x faster
        on multi-million-row candidate frames)."""
        if not len(recs):
            return recs.reset_index(drop=True)
        q_codes, _ = pd.factorize(recs[query_column], sort=True)
        ratings = recs[rating_column].to_numpy(np.float64)
        if np.all(np.diff(q_codes) >= 0):
            # frame already grouped by query (sparse-matmul outputs are):
            # per-segment argpartition beats a full lexsort several-x
            starts = np.r_[0, np.flatnonzero(np.diff(q_codes)) + 1, len(q_codes)]
            keep = []
            for lo, hi in zip(starts[:-1], starts[1:]):
                seg = ratings[lo:hi]
                if hi - lo > k:
                    top = np.argpartition(-seg, k - 1)[:k]
                else:
                    top = np.arange(hi - lo)
                keep.append(lo + top[np.argsort(-seg[top], kind="stable")])
            order = np.concatenate(keep)
            return recs.iloc[order].reset_index(drop=True)
        order = np.lexsort((-ratings, q_codes))  # stable: ties keep frame order
        sorted_q = q_codes[order]
        boundaries = np.flatnonzero(np.diff(sorted_q)) + 1
        starts = np.r_[0, boundaries]
        lengths = np.diff(np.r_[starts, len(sorted_q)])
        ranks = np.arange(len(sorted_q)) - np.repeat(starts, lengths)
        return recs.iloc[order[ranks < k]].reset_index(drop=True)

    def _predict_wrap(
        self,
        dataset,
        k: int,
        queries=None,
        items=None,
        filter_seen_items: bool = True,
        recs_file_path: Optional[str] = None,
    ) -> Optional[pd.DataFrame]:
        """Cold filtering -> model ``_predict`` -> seen filtering -> top-k
        (reference base_rec.py:258)."""
        queries_df = self._ids_frame(queries, self.query_column)
        if queries_df is None:
            queries_df = (
                pd.DataFrame({self.query_column: dataset.interactions[self.query_column].unique()})
                if dataset is not None
                else self.fit_queries
            )
        items_df = self._ids_frame(items, self.item_column)
        if items_df is None:
            items_df = self.fit_items
        queries_df, items_df = self._filter_cold_for_predict(queries_df, items_df)

        n_fetch = k
        if filter_seen_items and dataset is not None:
            inter = dataset.interactions
            seen_counts = (
                inter[inter[self.query_column].isin(set(queries_df[self.query_column]))]
                .groupby(self.query_column)
                .size()
            )
            max_seen = int(seen_counts.max()) if len(seen_counts) else 0
            n_fetch = k + max_seen

        recs = self._predict(dataset, n_fetch, queries_df, items_df, filter_seen_items)
        if filter_seen_items and dataset is not None:
            recs = self._filter_seen(recs, dataset.interactions, queries_df)
        recs = self._get_top_k(recs, self.query_column, self.rating_column, k)
        recs = recs[[self.query_column, self.item_column, self.rating_column]]
        if recs_file_path is not None:
            recs.to_parquet(recs_file_path, index=False)
            return None
        return recs

    def _predict(
        self, dataset, k: int, queries: pd.DataFrame, items: pd.DataFrame, filter_seen_items: bool = True
    ) -> pd.DataFrame:  # pragma: no cover
        raise NotImplementedError

    # -- matrix -> recs helper ---------------------------------------------------
    def _recs_from_scores(
        self, scores: np.ndarray, query_ids: np.ndarray, item_ids: np.ndarray, k: int
    ) -> pd.DataFrame:
        """Turn a dense [n_q, n_i] score matrix into a top-k recs frame."""
        k = min(k, scores.shape[1])
        if k <= 0 or scores.shape[0] == 0:
            return pd.DataFrame({self.query_column: [], self.item_column: [], self.rating_column: []})
        idx = np.argpartition(-scores, kth=k - 1, axis=1)[:, :k]
        top_scores = np.take_along_axis(scores, idx, axis=1)
        order = np.argsort(-top_scores, axis=1, kind="stable")
        idx = np.take_along_axis(idx, order, axis=1)
        top_scores = np.take_along_axis(top_scores, order, axis=1)
        return pd.DataFrame(
            {
                self.query_column: np.repeat(query_ids, k),
                self.item_column: item_ids[idx.ravel()],
                self.rating_column: top_scores.ravel().astype(np.float64),
            }
        )

    # -- predict_pairs ------------------------------------------------------------
    def _predict_pairs_wrap(self, pairs: pd.DataFrame, dataset=None, k: Optional[int] = None) -> pd.DataFrame:
        """Score explicit (query, item) pairs (reference base_rec.py:502) with
        a generic predict-then-join fallback (reference :569-604)."""
        queries = pairs[[self.query_column]].drop_duplicates()
        items = pairs[[self.item_column]].drop_duplicates()
        queries, items = self._filter_cold_for_predict(queries, items)
        pairs = pairs.merge(queries, on=self.query_column).merge(items, on=self.item_column)
        scored = self._predict_pairs(pairs, dataset)
        if k is not None:
            scored = self._get_top_k(scored, self.query_column, self.rating_column, k)
        return scored[[self.query_column, self.item_column, self.rating_column]]

    def _predict_pairs(self, pairs: pd.DataFrame, dataset=None) -> pd.DataFrame:
        """Fallback: full predict over the pair's items, then join."""
        queries = pairs[[self.query_column]].drop_duplicates()
        items = pairs[[self.item_column]].drop_duplicates()
        recs = self._predict(dataset, len(items), queries, items, filter_seen_items=False)
        return pairs.merge(recs, on=[self.query_column, self.item_column], how="left").fillna(
            {self.rating_column: -np.inf}
        )

    def _fit_predict(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        self._fit_wrap(dataset)
        return self._predict_wrap(dataset, k, queries, items, filter_seen_items, recs_file_path)

    # -- optimization hook (OptunaMixin-equivalent; see models/optimization) ---
    def optimize(self, *args, **kwargs):
        from replay_amd.models.optimization import optimize_model

        return optimize_model(self, *args, **kwargs)


class Recommender(BaseRecommender):
    """Standard interaction-only recommender (reference base_rec.py:926)."""

    def fit(self, dataset) -> None:
        self._fit_wrap(dataset)

    def predict(
        self,
        dataset,
        k: int,
        queries=None,
        items=None,
        filter_seen_items: bool = True,
        recs_file_path: Optional[str] = None,
    ) -> Optional[pd.DataFrame]:
        return self._predict_wrap(dataset, k, queries, items, filter_seen_items, recs_file_path)

    def predict_pairs(self, pairs: pd.DataFrame, dataset=None, recs_file_path=None, k=None):
        recs = self._predict_pairs_wrap(pairs, dataset, k)
        if recs_file_path is not None:
            recs.to_parquet(recs_file_path, index=False)
            return None
        return recs

    def fit_predict(
        self,
        dataset,
        k: int,
        queries=None,
        items=None,
        filter_seen_items: bool = True,
        recs_file_path: Optional[str] = None,
    ) -> Optional[pd.DataFrame]:
        return self._fit_predict(dataset, k, queries, items, filter_seen_items, recs_file_path)

    def get_features(self, ids: pd.DataFrame) -> Optional[pd.DataFrame]:
        """Return latent factors for ids, if the model has them
        (reference base_rec.py:1041)."""
        return None


class HybridRecommender(BaseRecommender):
    """Recommender that uses query/item features (reference base_rec.py:795)."""

    def fit(self, dataset) -> None:
        self._fit_wrap(dataset)

    def predict(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        return self._predict_wrap(dataset, k, queries, items, filter_seen_items, recs_file_path)

    def predict_pairs(self, pairs, dataset=None, recs_file_path=None, k=None):
        return self._predict_pairs_wrap(pairs, dataset, k)

    def fit_predict(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        return self._fit_predict(dataset, k, queries, items, filter_seen_items, recs_file_path)


class QueryRecommender(HybridRecommender):
    """Needs query features only (reference base_rec.py:1052)."""

    can_predict_cold_queries = True


class NonPersonalizedRecommender(Recommender):
    """Same recommendations for every query (reference base_rec.py:1143)."""

    can_predict_cold_queries = True
    can_predict_cold_items = True

    def __init__(self, add_cold_items: bool = True, cold_weight: float = 0.5) -> None:
        super().__init__()
        if not 0 < cold_weight <= 1:
            raise ValueError("cold_weight must be in (0, 1]")
        self.add_cold_items = add_cold_items
        self.cold_weight = cold_weight
        self.item_popularity: Optional[pd.DataFrame] = None

    @property
    def _dataframes(self):
        return {"item_popularity": self.item_popularity}

    def _fill_cold_ratings(self, items: pd.DataFrame) -> pd.DataFrame:
        """Join popularity; cold items get cold_weight * min rating
        (reference base_rec.py:1222 semantics)."""
        pop = self.item_popularity
        merged = items.merge(pop, on=self.item_column, how="left" if self.add_cold_items else "inner")
        if self.add_cold_items and merged[self.rating_column].isna().any():
            fill = (pop[self.rating_column].min() if len(pop) else 0.0) * self.cold_weight
            merged[self.rating_column] = merged[self.rating_column].fillna(fill)
        return merged

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        scored_items = self._fill_cold_ratings(items)
        scored_items = scored_items.nlargest(min(k, len(scored_items)), self.rating_column)
        recs = queries.merge(scored_items, how="cross")
        return recs


class ItemVectorModel:
    """Mixin: similar-item search over item vectors (reference base_rec.py:692)."""

    def _get_item_vectors(self) -> pd.DataFrame:  # pragma: no cover
        raise NotImplementedError

    def get_nearest_items(
        self, items: Iterable, k: int, metric: str = "cosine_similarity", candidates: Optional[Iterable] = None
    ) -> pd.DataFrame:
        """Top-k similar items by cosine similarity / dot product / euclidean
        distance (reference base_rec.py:709,740-792)."""
        vectors_df = self._get_item_vectors()
        item_col = self.item_column
        all_ids = vectors_df[item_col].to_numpy()
        mat = np.stack(vectors_df["item_vector"].to_numpy())
        if candidates is not None:
            cand_mask = np.isin(all_ids, np.asarray(list(candidates)))
        else:
            cand_mask = np.ones(len(all_ids), dtype=bool)
        cand_ids = all_ids[cand_mask]
        cand_mat = mat[cand_mask]
        out_rows = []
        query_items = np.asarray(list(items))
        for qi in query_items:
            pos = np.nonzero(all_ids == qi)[0]
            if not len(pos):
                continue
            v = mat[pos[0]]
            if metric == "cosine_similarity":
                denom = np.linalg.norm(cand_mat, axis=1) * (np.linalg.norm(v) + 1e-12) + 1e-12
                sim = cand_mat @ v / denom
            elif metric == "dot_product":
                sim = cand_mat @ v
            elif metric == "euclidean_distance_sim":
                sim = 1.0 / (1.0 + np.linalg.norm(cand_mat - v, axis=1))
            else:
                raise ValueError(f"Unknown metric {metric}")
            mask = cand_ids != qi
            ids_f, sim_f = cand_ids[mask], sim[mask]
            top = np.argsort(-sim_f, kind="stable")[:k]
            for j in top:
                out_rows.append((qi, ids_f[j], float(sim_f[j])))
        return pd.DataFrame(out_rows, columns=[item_col, "neighbour_item_id", "similarity"])
