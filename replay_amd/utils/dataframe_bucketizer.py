"""Bucketized-table optimization.

The reference DataframeBucketizer (replay/utils/dataframe_bucketizer.py:12)
pre-buckets a Spark table by query id to avoid shuffle in repeated joins.
The pandas counterpart sorts + indexes the frame by the bucket column so the
repeated per-query group lookups are O(1) slices.
"""

from __future__ import annotations

import pandas as pd


class DataframeBucketizer:
    def __init__(self, bucketing_key: str = "query_id", partition_num: int = 1, spark_warehouse_dir: str = "", table_name: str = "") -> None:
        self.bucketing_key = bucketing_key
        self.partition_num = partition_num

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        out = df.sort_values(self.bucketing_key, kind="stable").reset_index(drop=True)
        out.attrs["bucketed_by"] = self.bucketing_key
        return out

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False
