"""Backend type aliases and availability flags.

MI355X-native counterpart of the reference's conditional-import flag system
(reference: replay/utils/types.py:23-51).  The reference gates whole subsystems
on PYSPARK/TORCH/ANN/OPTUNA/OPENVINO availability; we do the same for the
subsystems that exist in this stack: torch+ROCm (the compute tier), polars
(optional tabular backend) and our HIP extension module.
"""

from typing import Iterable, Union

import pandas as pd


class MissingImport:
    """Placeholder type for gated-out backends (reference types.py:9)."""


class FeatureUnavailableError(Exception):
    """Raised when a gated subsystem is used without its dependency
    (reference types.py:15)."""


class FeatureUnavailableWarning(Warning):
    """Warned when a gated subsystem degrades (reference types.py:19)."""


IntOrList = Union[Iterable[int], int]
NumType = Union[int, float]

try:  # optional second tabular backend
    import polars as pl  # noqa: F401

    POLARS_AVAILABLE = True
    PolarsDataFrame = pl.DataFrame
except ImportError:  # pragma: no cover
    POLARS_AVAILABLE = False

    class PolarsDataFrame:  # type: ignore[no-redef]
        """Placeholder when polars is not installed."""


try:
    import torch  # noqa: F401

    TORCH_AVAILABLE = True
except ImportError:  # pragma: no cover
    TORCH_AVAILABLE = False

# Spark is intentionally unsupported in the MI355X build: the reference's Spark
# tier exists for CPU-cluster scale-out; our scale axis is GPUs over RCCL/xGMI.
PYSPARK_AVAILABLE = False


class SparkDataFrame:  # pragma: no cover
    """Placeholder type.  Spark is not supported by the MI355X build."""


PandasDataFrame = pd.DataFrame

if POLARS_AVAILABLE:
    DataFrameLike = Union[PandasDataFrame, PolarsDataFrame]
else:
    DataFrameLike = PandasDataFrame


def ROCM_AVAILABLE() -> bool:
    """True when running with a visible ROCm GPU."""
    if not TORCH_AVAILABLE:
        return False
    import torch

    return torch.cuda.is_available()


HIP_EXT_AVAILABLE = None  # resolved lazily by replay_amd.ops

# subsystems the reference gates that are N/A or unconditionally absent in
# the MI355X build (no JVM, no OpenVINO; ANN = our exact GPU brute force)
ANN_AVAILABLE = TORCH_AVAILABLE  # brute-force GPU index needs torch only
OPTUNA_AVAILABLE = False  # own search (models.optimization), no optuna dep
OPENVINO_AVAILABLE = False  # compiled inference = torch.jit instead
LIGHTFM_AVAILABLE = False
OBP_AVAILABLE = False  # own IPS/SNIPS estimators (experimental.scenarios)
