"""ULinUCB (reference replay/experimental/models/u_lin_ucb.py, 113 LoC):
user-side LinUCB — per-USER ridge state over item-factor features, scoring
mean + alpha * exploration."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from replay_amd.models.base_rec import Recommender


class ULinUCB(Recommender):
    def __init__(self, alpha: float = 1.0, rank: int = 16, seed: Optional[int] = None) -> None:
        super().__init__()
        self.alpha = alpha
        self.rank = rank
        self.seed = seed
        self._item_features: Optional[np.ndarray] = None
        self._theta: Optional[np.ndarray] = None
        self._a_inv: Optional[np.ndarray] = None

    @property
    def _init_args(self):
        return {"alpha": self.alpha, "rank": self.rank, "seed": self.seed}

    def _fit(self, dataset) -> None:
        # item features: SVD of the interaction matrix (self-contained arms)
        from scipy.sparse import csr_matrix
        from scipy.sparse.linalg import svds

        inter = dataset.interactions
        rows = inter[self.query_column].to_numpy(dtype=np.int64)
        cols = inter[self.item_column].to_numpy(dtype=np.int64)
        vals = (
            inter[self.rating_column].to_numpy(dtype=np.float64)
            if self.rating_column in inter.columns
            else np.ones(len(inter))
        )
        mat = csr_matrix((vals, (rows, cols)), shape=(self._query_dim_size, self._item_dim_size))
        rank = min(self.rank, min(mat.shape) - 1)
        _, s, vt = svds(mat.asfptype(), k=max(1, rank))
        self._item_features = (vt.T * s).astype(np.float64)  # [I, d]
        d = self._item_features.shape[1]
        # per-user ridge regression over interacted items
        theta = np.zeros((self._query_dim_size, d))
        a_inv = np.tile(np.eye(d), (self._query_dim_size, 1, 1))
        for u in range(self._query_dim_size):
            mask = rows == u
            if not mask.any():
                continue
            X = self._item_features[cols[mask]]
            r = vals[mask]
            A = np.eye(d) + X.T @ X
            Ainv = np.linalg.inv(A)
            theta[u] = Ainv @ (X.T @ r)
            a_inv[u] = Ainv
        self._theta, self._a_inv = theta, a_inv

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        X = self._item_features[i_ids]  # [I, d]
        mean = self._theta[q_ids] @ X.T  # [Q, I]
        expl = np.zeros_like(mean)
        for qi, u in enumerate(q_ids):
            xa = X @ self._a_inv[u]  # [I, d]
            expl[qi] = np.sqrt(np.maximum((xa * X).sum(-1), 0.0))
        scores = mean + self.alpha * expl
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))
