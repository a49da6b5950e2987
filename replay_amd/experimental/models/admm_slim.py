"""ADMM SLIM (reference replay/experimental/models/admm_slim.py, 257 LoC):
item-item weight matrix solved by ADMM with L1+L2 regularization and
zero-diagonal constraint (Steck et al. 2020 closed-form updates)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
from scipy.sparse import csr_matrix

from replay_amd.models.knn import NeighbourRec


class ADMMSLIM(NeighbourRec):
    def __init__(
        self,
        lambda_1: float = 5.0,
        lambda_2: float = 5000.0,
        seed: Optional[int] = None,
        num_iterations: int = 50,
        rho: Optional[float] = None,
    ) -> None:
        super().__init__()
        if lambda_1 < 0 or lambda_2 <= 0:
            raise ValueError("lambda_1 must be >= 0, lambda_2 > 0")
        self.lambda_1 = lambda_1
        self.lambda_2 = lambda_2
        self.seed = seed
        self.num_iterations = num_iterations
        self.rho = rho if rho is not None else lambda_2

    @property
    def _init_args(self):
        return {
            "lambda_1": self.lambda_1,
            "lambda_2": self.lambda_2,
            "seed": self.seed,
            "num_iterations": self.num_iterations,
        }

    _search_space = {
        "lambda_1": {"type": "loguniform", "args": [1e-9, 50]},
        "lambda_2": {"type": "loguniform", "args": [1e-9, 5000]},
    }

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        rows = inter[self.query_column].to_numpy(dtype=np.int64)
        cols = inter[self.item_column].to_numpy(dtype=np.int64)
        data = np.ones(len(inter))
        X = csr_matrix((data, (rows, cols)), shape=(self._query_dim_size, self._item_dim_size))
        G = (X.T @ X).toarray()  # gram, [I, I]
        n = G.shape[0]
        rho = self.rho
        P = np.linalg.inv(G + (self.lambda_2 + rho) * np.eye(n))
        B_aux = P @ G
        gamma = np.zeros((n, n))
        C = np.zeros((n, n))
        for _ in range(self.num_iterations):
            B = B_aux + P @ (rho * C - gamma)
            # zero-diagonal correction
            diag = np.diag(B) / np.maximum(np.diag(P), 1e-12)
            B -= P * diag[None, :]
            # soft-threshold for L1 + projection
            raw = B + gamma / rho
            C = np.sign(raw) * np.maximum(np.abs(raw) - self.lambda_1 / rho, 0.0)
            np.fill_diagonal(C, 0.0)
            gamma += rho * (B - C)
        nz = np.nonzero(C)
        self.similarity = pd.DataFrame(
            {"item_idx_one": nz[0], "item_idx_two": nz[1], "similarity": C[nz]}
        )
