"""ANN mixin: swap a model's exhaustive predict for an index query.

Parity with reference ANNMixin (replay/models/extensions/ann/ann_mixin.py:26):
overrides ``_fit_wrap`` to build an index from the model's item vectors after
fitting and ``_predict_wrap`` to query it (with seen-item filtering, the
reference's filter-seen inferer variants).  Usable with any ItemVectorModel
(ALSWrap, Word2VecRec) the way the reference wires ItemKNN/Word2Vec/SLIM
(knn.py:18-21, word2vec.py:27-56).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from .index import BruteForceIndex, IndexParams


class ANNMixin:
    """Mix into a Recommender that implements ``_get_item_vectors`` and a
    per-query vector method ``_get_query_vectors(queries, dataset)``."""

    index_params: Optional[IndexParams] = None
    _index: Optional[BruteForceIndex] = None

    def _init_ann(self, index_params: Optional[IndexParams] = None) -> None:
        self.index_params = index_params or IndexParams()

    def _build_ann_index(self) -> None:
        vectors_df = self._get_item_vectors()
        order = np.argsort(vectors_df[self.item_column].to_numpy())
        mat = np.stack(vectors_df["item_vector"].to_numpy())[order]
        self._index = BruteForceIndex(self.index_params).build(mat)

    def _fit_wrap(self, dataset) -> None:
        super()._fit_wrap(dataset)
        if self.index_params is not None:
            self._build_ann_index()

    def _get_query_vectors(self, queries: pd.DataFrame, dataset) -> np.ndarray:
        if hasattr(self, "user_factors"):
            return self.user_factors[queries[self.query_column].to_numpy(dtype=np.int64)]
        if hasattr(self, "_query_vectors"):
            return self._query_vectors(dataset, queries[self.query_column].to_numpy())
        raise NotImplementedError

    def _predict_wrap(self, dataset, k, queries=None, items=None, filter_seen_items=True, recs_file_path=None):
        if self._index is None:
            return super()._predict_wrap(dataset, k, queries, items, filter_seen_items, recs_file_path)
        queries_df = self._ids_frame(queries, self.query_column)
        if queries_df is None:
            queries_df = (
                pd.DataFrame({self.query_column: dataset.interactions[self.query_column].unique()})
                if dataset is not None
                else self.fit_queries
            )
        queries_df, _ = self._filter_cold_for_predict(queries_df, self.fit_items)
        qv = self._get_query_vectors(queries_df, dataset)
        filter_items = None
        if filter_seen_items and dataset is not None:
            seen = (
                dataset.interactions.groupby(self.query_column)[self.item_column].apply(set).to_dict()
            )
            filter_items = [seen.get(q) for q in queries_df[self.query_column]]
        scores, ids = self._index.search(qv, k, filter_items)
        q_ids = queries_df[self.query_column].to_numpy()
        recs = pd.DataFrame(
            {
                self.query_column: np.repeat(q_ids, ids.shape[1]),
                self.item_column: ids.ravel(),
                self.rating_column: scores.ravel().astype(np.float64),
            }
        )
        recs = recs[np.isfinite(recs[self.rating_column])]
        if recs_file_path is not None:
            recs.to_parquet(recs_file_path, index=False)
            return None
        return recs
