"""Time-smoothing weights (reference replay/utils/time.py:10 get_item_recency)."""

from __future__ import annotations

import numpy as np
import pandas as pd


def smoothe_time(
    log: pd.DataFrame,
    decay: float = 30.0,
    limit: float = 0.1,
    kind: str = "exp",
    timestamp_column: str = "timestamp",
) -> pd.DataFrame:
    """Add a ``relevance`` in (0, 1] decaying with interaction age."""
    df = log.copy()
    ts = df[timestamp_column]
    if pd.api.types.is_datetime64_any_dtype(ts):
        seconds = ts.astype("int64") / 10**9
    else:
        seconds = ts.astype("float64")
    days_old = (seconds.max() - seconds) / 86400.0
    if kind == "power":
        weight = (days_old + 1.0) ** (np.log(limit) / np.log(decay + 1))
    elif kind == "exp":
        weight = np.exp(days_old * np.log(limit) / decay)
    elif kind == "linear":
        weight = np.clip(1.0 + days_old * (limit - 1.0) / decay, limit, 1.0)
    else:
        raise ValueError("kind must be power/exp/linear")
    df["relevance"] = weight
    return df


def get_item_recency(
    log: pd.DataFrame,
    decay: float = 30.0,
    limit: float = 0.1,
    kind: str = "exp",
    timestamp_column: str = "timestamp",
    item_column: str = "item_id",
) -> pd.DataFrame:
    """Per-item recency weight computed at the item's mean timestamp."""
    items = log.groupby(item_column, as_index=False)[timestamp_column].mean()
    return smoothe_time(items, decay, limit, kind, timestamp_column)
