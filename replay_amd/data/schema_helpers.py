"""Frame-schema helper (the pandas counterpart of reference
replay/data/spark_schema.py:7 get_schema)."""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np


def get_schema(
    query_column: str = "query_id",
    item_column: str = "item_id",
    timestamp_column: Optional[str] = "timestamp",
    rating_column: Optional[str] = "rating",
) -> Dict[str, np.dtype]:
    """Column -> numpy dtype mapping for an interactions frame."""
    schema = {query_column: np.dtype("int64"), item_column: np.dtype("int64")}
    if timestamp_column:
        schema[timestamp_column] = np.dtype("int64")
    if rating_column:
        schema[rating_column] = np.dtype("float64")
    return schema
