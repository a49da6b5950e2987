from .types import (
    POLARS_AVAILABLE,
    PYSPARK_AVAILABLE,
    TORCH_AVAILABLE,
    DataFrameLike,
    PandasDataFrame,
    PolarsDataFrame,
    SparkDataFrame,
)

__all__ = [
    "POLARS_AVAILABLE",
    "PYSPARK_AVAILABLE",
    "TORCH_AVAILABLE",
    "DataFrameLike",
    "PandasDataFrame",
    "PolarsDataFrame",
    "SparkDataFrame",
]
