"""Frame helpers (the pandas-native counterparts of reference
replay/utils/spark_utils.py: get_top_k_recs:156, filter_cold:724,
convert2spark:78, fallback:480)."""

from __future__ import annotations

from typing import Optional, Tuple

import pandas as pd


def get_top_k(
    frame: pd.DataFrame,
    partition_by: str,
    order_by: Tuple[str, bool],
    k: int,
) -> pd.DataFrame:
    """Top-k rows per partition ordered by (column, ascending)."""
    column, ascending = order_by
    out = frame.sort_values([partition_by, column], ascending=[True, ascending], kind="stable")
    return out.groupby(partition_by, sort=False).head(k).reset_index(drop=True)


def get_top_k_recs(recs: pd.DataFrame, k: int, query_column: str = "query_id", rating_column: str = "rating") -> pd.DataFrame:
    """Top-k recommendations per query by descending rating
    (reference spark_utils.py:156)."""
    return get_top_k(recs, query_column, (rating_column, False), k)


def filter_cold(
    df: Optional[pd.DataFrame],
    warm_df: pd.DataFrame,
    col_name: str,
) -> Tuple[int, Optional[pd.DataFrame]]:
    """Drop rows whose id is not in warm_df; returns (n_dropped, filtered)
    (reference spark_utils.py:724)."""
    if df is None:
        return 0, None
    warm = set(warm_df[col_name])
    mask = df[col_name].isin(warm)
    return int((~mask).sum()), df[mask]


def fallback(
    base: pd.DataFrame,
    fill: pd.DataFrame,
    k: int,
    query_column: str = "query_id",
    item_column: str = "item_id",
    rating_column: str = "rating",
) -> pd.DataFrame:
    """Merge main recs with fallback recs, fallback shifted below the main
    minimum (reference spark_utils.py:480)."""
    from replay_amd.scenarios.fallback import Fallback

    merged = Fallback._merge_recs(base, fill, query_column, rating_column)
    return get_top_k_recs(merged, k, query_column, rating_column)


def convert2pandas(df) -> pd.DataFrame:
    """Backend conversion entry point (the reference's convert2spark
    counterpart: in this stack everything converges on pandas/Arrow)."""
    if isinstance(df, pd.DataFrame):
        return df
    try:  # polars
        import polars as pl

        if isinstance(df, pl.DataFrame):  # pragma: no cover
            return df.to_pandas()
    except ImportError:
        pass
    import pyarrow as pa

    if isinstance(df, pa.Table):
        return df.to_pandas()
    raise TypeError(f"Cannot convert {type(df)} to pandas")
