"""Canonical interaction-frame dtypes (reference replay/data/
spark_schema.py:7 ``get_schema`` — there a Spark StructType; here the
pandas/numpy dtype mapping used to validate and cast interaction logs)."""

from __future__ import annotations

from typing import Dict

import numpy as np


def get_schema(
    query_column: str = "query_id",
    item_column: str = "item_id",
    timestamp_column: str = "timestamp",
    rating_column: str = "rating",
    has_timestamp: bool = True,
    has_rating: bool = True,
) -> Dict[str, np.dtype]:
    """Column -> numpy dtype for the canonical interaction log layout."""
    schema: Dict[str, np.dtype] = {
        query_column: np.dtype("int64"),
        item_column: np.dtype("int64"),
    }
    if has_timestamp:
        schema[timestamp_column] = np.dtype("int64")
    if has_rating:
        schema[rating_column] = np.dtype("float64")
    return schema
