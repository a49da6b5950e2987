"""Distribution helpers (reference replay/utils/distributions.py:11)."""

from __future__ import annotations

import pandas as pd


def item_distribution(
    log: pd.DataFrame,
    recommendations: pd.DataFrame,
    k: int,
    query_column: str = "query_id",
    item_column: str = "item_id",
    rating_column: str = "rating",
) -> pd.DataFrame:
    """Item popularity in the log vs its top-k recommendation count."""
    pop = log.groupby(item_column)[query_column].nunique().rename("user_count").reset_index()
    recs = recommendations.sort_values(rating_column, ascending=False, kind="stable")
    topk = recs.groupby(query_column, sort=False).head(k)
    rec_counts = topk.groupby(item_column).size().rename("rec_count").reset_index()
    out = pop.merge(rec_counts, on=item_column, how="outer").fillna(0)
    out["rec_count"] = out["rec_count"].astype(int)
    return out.sort_values(item_column).reset_index(drop=True)
