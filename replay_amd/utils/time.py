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
    """Weigh the ``relevance`` column by a time-decay factor (reference
    utils/time.py:114): an interaction aged ``decay`` days weighs 0.5;
    weights floor at ``limit``; existing relevance is MULTIPLIED.
    ``power``: (age+1)^(ln 0.5 / ln decay); ``exp``: 0.5^(age/decay);
    ``linear``: 1 - 0.5/decay * age."""
    df = log.copy()
    ts = df[timestamp_column]
    if not pd.api.types.is_numeric_dtype(ts):
        ts = pd.to_datetime(ts)
        seconds = ts.astype("int64") / 10**9
        df[timestamp_column] = ts
    else:
        seconds = ts.astype("float64")
    age = (seconds.max() - seconds) / 86400.0
    if kind == "power":
        weight = (age + 1.0) ** (np.log(0.5) / np.log(decay))
    elif kind == "exp":
        weight = np.exp(np.log(0.5) / decay) ** age
    elif kind == "linear":
        weight = 1.0 - (0.5 / decay) * age
    else:
        raise ValueError(f"parameter kind must be one of [power, exp, linear], got {kind}")
    weight = np.maximum(weight, limit)
    if "relevance" in df.columns:
        df["relevance"] = df["relevance"] * weight
    else:
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
    """Per-item recency weight computed at the item's mean timestamp
    (reference utils/time.py:10)."""
    df = log.copy()
    ts = df[timestamp_column]
    if not pd.api.types.is_numeric_dtype(ts) and not pd.api.types.is_datetime64_any_dtype(ts):
        df[timestamp_column] = pd.to_datetime(ts)
    items = df.groupby(item_column, as_index=False)[timestamp_column].mean()
    if "relevance" in log.columns:
        items = items.merge(
            log.groupby(item_column, as_index=False)["relevance"].mean(), on=item_column
        )
    return smoothe_time(items, decay, limit, kind, timestamp_column)
