"""SLIM: sparse linear item-item model via per-item ElasticNet.

Parity with reference SLIM (replay/models/slim.py:20): per-item-column
ElasticNet regression over the user-item matrix (reference slim.py:106 uses a
Spark ``applyInPandas``; here sklearn ElasticNet column-by-column), positive
coefficients only, diagonal excluded.
"""

from __future__ import annotations

from typing import Optional

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
        model = ElasticNet(
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
        out_rows, out_cols, out_data = [], [], []
        for j in range(mat.shape[1]):
            y = np.asarray(mat[:, j].todense()).ravel()
            if not y.any():
                continue
            start, end = mat.indptr[j], mat.indptr[j + 1]
            backup = mat.data[start:end].copy()
            mat.data[start:end] = 0.0  # exclude the target column
            model.fit(mat, y)
            mat.data[start:end] = backup
            coef = model.sparse_coef_.tocoo()
            for i, v in zip(coef.col, coef.data):
                if v > 0 and i != j:
                    out_rows.append(i)
                    out_cols.append(j)
                    out_data.append(v)
        self.similarity = pd.DataFrame(
            {"item_idx_one": out_rows, "item_idx_two": out_cols, "similarity": out_data}
        )
