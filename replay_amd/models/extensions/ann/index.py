"""Nearest-neighbour index.

MI355X replacement for the reference ANN extension's hnswlib/nmslib backends
(replay/models/extensions/ann/index_builders/*, index_inferers/*,
index_stores/*): on a 288 GB-HBM GPU a brute-force top-K GEMM over the full
item-factor matrix is exact AND faster than graph indices at recommender
catalog sizes (SURVEY §2.6 note), so the default "index" is a dense matrix
scored by hipBLASLt GEMM + torch.topk (chunked over items).  The index
save/load surface matches the reference's shared-disk store semantics.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional, Tuple

import numpy as np


@dataclass
class IndexParams:
    """Parameter entity (reference entities/*.py)."""

    space: str = "ip"  # ip | cosine | l2
    chunk_items: int = 1_000_000
    device: Optional[str] = None
    extra: dict = field(default_factory=dict)


class BruteForceIndex:
    def __init__(self, params: Optional[IndexParams] = None) -> None:
        self.params = params or IndexParams()
        self._vectors: Optional[np.ndarray] = None
        self._norms: Optional[np.ndarray] = None

    def build(self, item_vectors: np.ndarray) -> "BruteForceIndex":
        self._vectors = np.ascontiguousarray(item_vectors, dtype=np.float32)
        if self.params.space == "cosine":
            self._norms = np.linalg.norm(self._vectors, axis=1) + 1e-12
        return self

    @property
    def n_items(self) -> int:
        return 0 if self._vectors is None else self._vectors.shape[0]

    def search(
        self, queries: np.ndarray, k: int, filter_items: Optional[list] = None
    ) -> Tuple[np.ndarray, np.ndarray]:
        """Exact top-k: returns (scores [B, k], ids [B, k]).
        filter_items: per-query lists of item ids to exclude (seen items)."""
        import torch

        device = self.params.device or ("cuda" if torch.cuda.is_available() else "cpu")
        q = torch.from_numpy(np.ascontiguousarray(queries, dtype=np.float32)).to(device)
        v = torch.from_numpy(self._vectors).to(device)
        if self.params.space == "cosine":
            v = v / torch.from_numpy(self._norms).to(device)[:, None]
            q = q / (q.norm(dim=1, keepdim=True) + 1e-12)
        if self.params.space == "l2":
            scores = -torch.cdist(q, v)
        else:
            scores = q @ v.T
        if filter_items is not None:
            for i, items in enumerate(filter_items):
                if items is not None and len(items):
                    scores[i, torch.as_tensor(list(items), device=device)] = float("-inf")
        k = min(k, scores.shape[1])
        top_scores, top_ids = torch.topk(scores, k, dim=1)
        return top_scores.cpu().numpy(), top_ids.cpu().numpy()

    # -- persistence (reference index_stores shared-disk semantics) ------------
    def save(self, path) -> None:
        base = Path(path)
        base.mkdir(parents=True, exist_ok=True)
        np.savez(base / "index.npz", vectors=self._vectors, space=self.params.space)

    @classmethod
    def load(cls, path) -> "BruteForceIndex":
        data = np.load(Path(path) / "index.npz", allow_pickle=True)
        index = cls(IndexParams(space=str(data["space"])))
        index.build(data["vectors"])
        return index
