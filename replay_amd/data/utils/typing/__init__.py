from .dtype import numpy_to_pyarrow, numpy_to_torch, pyarrow_to_numpy, torch_to_numpy

__all__ = ["numpy_to_pyarrow", "numpy_to_torch", "pyarrow_to_numpy", "torch_to_numpy"]
