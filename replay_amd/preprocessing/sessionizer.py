"""Session-id assignment by time gap (reference replay/preprocessing/sessionizer.py:11)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd


class Sessionizer:
    """Assign session ids: a new session starts when the gap between a user's
    consecutive interactions exceeds ``session_gap`` (seconds).  Optionally
    filters sessions by length bounds."""

    def __init__(
        self,
        user_column: str = "user_id",
        time_column: str = "timestamp",
        session_column: str = "session_id",
        session_gap: float = 86400.0,
        min_inter_per_session: Optional[int] = None,
        max_inter_per_session: Optional[int] = None,
        time_column_format: str = "yyyy-MM-dd HH:mm:ss",
    ) -> None:
        self.user_column = user_column
        self.time_column = time_column
        self.session_column = session_column
        self.session_gap = session_gap
        self.min_inter_per_session = min_inter_per_session
        self.max_inter_per_session = max_inter_per_session

    def transform(self, interactions: pd.DataFrame) -> pd.DataFrame:
        df = interactions.sort_values([self.user_column, self.time_column], kind="stable").copy()
        ts = df[self.time_column]
        if pd.api.types.is_datetime64_any_dtype(ts):
            seconds = ts.astype("int64") // 10**9
        else:
            seconds = ts.astype("int64")
        new_user = df[self.user_column].ne(df[self.user_column].shift())
        gap = seconds.diff().fillna(0)
        new_session = new_user | (gap >= self.session_gap)  # reference: gap >= session_gap splits
        df[self.session_column] = np.cumsum(new_session.to_numpy()).astype(np.int64) - 1
        if self.min_inter_per_session is not None or self.max_inter_per_session is not None:
            sizes = df.groupby(self.session_column)[self.session_column].transform("size")
            mask = pd.Series(True, index=df.index)
            if self.min_inter_per_session is not None:
                mask &= sizes >= self.min_inter_per_session
            if self.max_inter_per_session is not None:
                mask &= sizes <= self.max_inter_per_session
            df = df[mask]
        return df.reindex(interactions.index.intersection(df.index))
