"""Association-rules item-to-item recommender.

Parity with reference AssociationRulesItemRec
(replay/models/association_rules.py:17): pair confidence / lift / confidence-gain
metrics over co-occurring item pairs inside sessions, top-k neighbour storage,
predict via the NeighbourRec similarity join.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd
from scipy.sparse import csr_matrix

from .knn import NeighbourRec


class AssociationRulesItemRec(NeighbourRec):
    def __init__(
        self,
        session_column: Optional[str] = None,
        min_item_count: int = 5,
        min_pair_count: int = 5,
        num_neighbours: Optional[int] = 1000,
        use_rating: bool = False,
        similarity_metric: str = "confidence",
    ) -> None:
        super().__init__()
        if similarity_metric not in ("confidence", "lift", "confidence_gain"):
            raise ValueError("similarity_metric must be confidence/lift/confidence_gain")
        self.session_column = session_column
        self.min_item_count = min_item_count
        self.min_pair_count = min_pair_count
        self.num_neighbours = num_neighbours
        self.use_rating = use_rating
        self.similarity_metric = similarity_metric

    @property
    def _init_args(self):
        return {
            "session_column": self.session_column,
            "min_item_count": self.min_item_count,
            "min_pair_count": self.min_pair_count,
            "num_neighbours": self.num_neighbours,
            "use_rating": self.use_rating,
            "similarity_metric": self.similarity_metric,
        }

    _search_space = {
        "min_item_count": {"type": "int", "args": [3, 10]},
        "min_pair_count": {"type": "int", "args": [3, 10]},
        "num_neighbours": {"type": "int", "args": [300, 2000]},
        "similarity_metric": {"type": "categorical", "args": ["confidence", "lift"]},
    }

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        group_col = self.session_column or self.query_column
        sessions = inter[group_col].astype("category").cat.codes.to_numpy()
        item_ids = inter[self.item_column].to_numpy(dtype=np.int64)
        n_sessions = int(sessions.max()) + 1 if len(sessions) else 0
        mat = csr_matrix(
            (np.ones(len(inter)), (sessions, item_ids)),
            shape=(n_sessions, self._item_dim_size),
        )
        mat.data[:] = 1.0  # binarize multiplicities
        item_count = np.asarray(mat.sum(axis=0)).ravel()
        frequent = item_count >= self.min_item_count

        pair = (mat.T @ mat).tocoo()
        rows, cols, data = pair.row, pair.col, pair.data
        mask = (rows != cols) & (data >= self.min_pair_count) & frequent[rows] & frequent[cols]
        rows, cols, data = rows[mask], cols[mask], data[mask]

        confidence = data / item_count[rows]
        consequent_share = item_count[cols] / max(1, n_sessions)
        lift = confidence / np.maximum(consequent_share, 1e-12)
        confidence_gain = confidence / np.maximum(
            (item_count[cols] - data) / np.maximum(n_sessions - item_count[rows], 1), 1e-12
        )
        sim = {"confidence": confidence, "lift": lift, "confidence_gain": confidence_gain}[
            self.similarity_metric
        ]
        df = pd.DataFrame(
            {
                "item_idx_one": rows,
                "item_idx_two": cols,
                "similarity": sim,
                "confidence": confidence,
                "lift": lift,
                "confidence_gain": confidence_gain,
            }
        )
        df = df.sort_values(["item_idx_one", "similarity"], ascending=[True, False], kind="stable")
        if self.num_neighbours is not None:
            df = df.groupby("item_idx_one", sort=False).head(self.num_neighbours)
        self.similarity = df.reset_index(drop=True)

    def get_nearest_items(self, items, k: int, metric: Optional[str] = "lift") -> pd.DataFrame:
        sim = self.similarity
        metric = metric or "similarity"
        sel = sim[sim["item_idx_one"].isin(set(items))].copy()
        sel = sel.sort_values(["item_idx_one", metric], ascending=[True, False], kind="stable")
        return sel.groupby("item_idx_one", sort=False).head(k).reset_index(drop=True)
