"""Interactions -> scipy CSR matrix (reference replay/preprocessing/converter.py:10)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
from scipy.sparse import csr_matrix


class CSRConverter:
    """Build a query x item CSR matrix from an interaction frame."""

    def __init__(
        self,
        first_dim_column: str = "user_id",
        second_dim_column: str = "item_id",
        data_column: Optional[str] = None,
        row_count: Optional[int] = None,
        column_count: Optional[int] = None,
        allow_collect_to_master: bool = False,
    ) -> None:
        self.first_dim_column = first_dim_column
        self.second_dim_column = second_dim_column
        self.data_column = data_column
        self.row_count = row_count
        self.column_count = column_count

    def transform(self, interactions: pd.DataFrame) -> csr_matrix:
        rows = interactions[self.first_dim_column].to_numpy(dtype=np.int64)
        cols = interactions[self.second_dim_column].to_numpy(dtype=np.int64)
        if self.data_column is not None:
            data = interactions[self.data_column].to_numpy(dtype=np.float64)
        else:
            data = np.ones(len(interactions), dtype=np.float64)
        n_rows = self.row_count if self.row_count is not None else (int(rows.max()) + 1 if len(rows) else 0)
        n_cols = self.column_count if self.column_count is not None else (int(cols.max()) + 1 if len(cols) else 0)
        return csr_matrix((data, (rows, cols)), shape=(n_rows, n_cols))
