from .columns import Array1DColumn, Array2DColumn, NamedColumns, NumericColumn, mask_name
from .info import DistributedInfo, ReplicasInfo, WorkerInfo
from .metadata import (
    ColumnMetadata,
    Metadata,
    get_1d_array_columns,
    get_2d_array_columns,
    get_numeric_columns,
    get_padding,
    get_shape,
)
from .parquet_dataset import FixedBatchSizeDataset, ParquetDataset
from .parquet_module import ParquetModule
from .partitioning import Partitioning

DEFAULT_REPLICAS_INFO = ReplicasInfo()
ReplicasInfoProtocol = ReplicasInfo  # structural: anything with curr/num replicas

__all__ = [
    "Array1DColumn",
    "Array2DColumn",
    "ColumnMetadata",
    "Metadata",
    "NamedColumns",
    "NumericColumn",
    "mask_name",
    "get_1d_array_columns",
    "get_2d_array_columns",
    "get_numeric_columns",
    "get_padding",
    "get_shape",
    "DistributedInfo",
    "DEFAULT_REPLICAS_INFO",
    "ReplicasInfo",
    "ReplicasInfoProtocol",
    "WorkerInfo",
    "FixedBatchSizeDataset",
    "ParquetDataset",
    "ParquetModule",
    "Partitioning",
]
