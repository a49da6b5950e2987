"""dtype bridges between torch, numpy and pyarrow (reference
replay/data/utils/typing/dtype.py): conversions go through a zero-element
exemplar so every dtype the libraries can exchange round-trips without a
hand-maintained table."""

from __future__ import annotations

from functools import lru_cache

import numpy as np
import torch

try:
    import pyarrow as pa
except ImportError:  # pragma: no cover
    pa = None


@lru_cache(maxsize=None)
def _torch_to_numpy(dtype: torch.dtype) -> np.dtype:
    return torch.zeros((), dtype=dtype).numpy().dtype


def torch_to_numpy(dtype: torch.dtype) -> np.dtype:
    return _torch_to_numpy(dtype)


def numpy_to_torch(dtype) -> torch.dtype:
    return torch.from_numpy(np.zeros((), dtype=np.dtype(dtype))).dtype


def numpy_to_pyarrow(dtype):
    if pa is None:  # pragma: no cover
        raise ImportError("pyarrow is not available")
    return pa.from_numpy_dtype(np.dtype(dtype))


def pyarrow_to_numpy(dtype) -> np.dtype:
    if pa is None:  # pragma: no cover
        raise ImportError("pyarrow is not available")
    return np.dtype(dtype.to_pandas_dtype())
