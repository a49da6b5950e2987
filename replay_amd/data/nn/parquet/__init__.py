from .columns import Array1DColumn, Array2DColumn, NamedColumns, NumericColumn, mask_name
from .info import DistributedInfo, ReplicasInfo, WorkerInfo
from .parquet_dataset import FixedBatchSizeDataset, ParquetDataset
from .parquet_module import ParquetModule
from .partitioning import Partitioning

__all__ = [
    "Array1DColumn",
    "Array2DColumn",
    "NamedColumns",
    "NumericColumn",
    "mask_name",
    "DistributedInfo",
    "ReplicasInfo",
    "WorkerInfo",
    "FixedBatchSizeDataset",
    "ParquetDataset",
    "ParquetModule",
    "Partitioning",
]
