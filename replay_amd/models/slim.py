"""SLIM: sparse linear item-item model via per-item ElasticNet.

Parity with reference SLIM (replay/models/slim.py:20): per-item-column
ElasticNet regression over the user-item matrix (reference slim.py:106 uses a
Spark ``applyInPandas``; here sklearn ElasticNet column-by-column), positive
coefficients only, diagonal excluded.
"""

from __future__ import annotations

from typing import Optional

import os

import numpy as np
import pandas as pd
from scipy.sparse import csr_matrix

from .knn import NeighbourRec


class SLIM(NeighbourRec):
    def __init__(self, beta: float = 0.01, lambda_: float = 0.01, seed: Optional[int] = None, allow_collect_to_master: bool = False) -> None:
        super().__init__()
        if beta < 0 or lambda_ < 0 or (beta == 0 and lambda_ == 0):
            raise ValueError("beta and lambda_ must be non-negative and not both zero")
        self.beta = beta
        self.lambda_ = lambda_
        self.seed = seed

    @property
    def _init_args(self):
        return {"beta": self.beta, "lambda_": self.lambda_, "seed": self.seed}

    _search_space = {
        "beta": {"type": "loguniform", "args": [1e-6, 5]},
        "lambda_": {"type": "loguniform", "args": [1e-6, 2]},
    }

    def _fit(self, dataset) -> None:
        from sklearn.linear_model import ElasticNet

        inter = dataset.interactions
        rows = inter[self.query_column].to_numpy(dtype=np.int64)
        cols = inter[self.item_column].to_numpy(dtype=np.int64)
        if self.rating_column in inter.columns:
            data = inter[self.rating_column].to_numpy(dtype=np.float64)
        else:
            data = np.ones(len(inter))
        mat = csr_matrix((data, (rows, cols)), shape=(self._query_dim_size, self._item_dim_size)).tocsc()

        alpha = self.beta + self.lambda_
        l1_ratio = self.lambda_ / alpha
        params = dict(
            alpha=alpha,
            l1_ratio=l1_ratio,
            positive=True,
            fit_intercept=False,
            copy_X=False,
            precompute=True,
            selection="random",
            max_iter=100,
            tol=1e-4,
            random_state=self.seed,
        )

        def fit_columns(col_range):
            model = ElasticNet(**params)
            local = mat.copy()
            rows_o, cols_o, data_o = [], [], []
            for j in col_range:
                start, end = local.indptr[j], local.indptr[j + 1]
                if start == end:
                    continue
                y = np.zeros(local.shape[0])
                y[local.indices[start:end]] = local.data[start:end]
                backup = local.data[start:end].copy()
                local.data[start:end] = 0.0  # exclude the target column
                model.fit(local, y)
                local.data[start:end] = backup
                coef = model.sparse_coef_.tocoo()
                keep = (coef.data > 0) & (coef.col != j)
                rows_o.append(coef.col[keep].astype(np.int64))
                cols_o.append(np.full(int(keep.sum()), j, dtype=np.int64))
                data_o.append(coef.data[keep])
            cat = lambda parts, dt: np.concatenate(parts) if parts else np.array([], dtype=dt)
            return cat(rows_o, np.int64), cat(cols_o, np.int64), cat(data_o, np.float64)

        # one independent ElasticNet per item column: parallelize across
        # processes (the reference distributes the same loop over Spark
        # executors, replay/models/slim.py)
        try:
            from joblib import Parallel, delayed

            n_jobs = min(8, max(1, (os.cpu_count() or 1)))
            chunks = np.array_split(np.arange(mat.shape[1]), n_jobs * 4)
            results = Parallel(n_jobs=n_jobs)(delayed(fit_columns)(c) for c in chunks if len(c))
        except ImportError:  # pragma: no cover
            results = [fit_columns(np.arange(mat.shape[1]))]
        self.similarity = pd.DataFrame(
            {
                "item_idx_one": np.concatenate([r[0] for r in results]),
                "item_idx_two": np.concatenate([r[1] for r in results]),
                "similarity": np.concatenate([r[2] for r in results]),
            }
        )
