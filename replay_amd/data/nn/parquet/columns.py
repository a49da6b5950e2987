"""Arrow column -> torch tensor decoders.

Parity with reference replay/data/nn/parquet/impl/: NumericColumn
(numeric_column.py), Array1DColumn (array_1d_column.py — ragged list ->
padded [B, L] + bool mask), Array2DColumn (array_2d_column.py:22 — nested
lists -> [B, L, W] via offsets + mask), NamedColumns (named_columns.py),
mask naming (masking.py:12: ``<name>_mask``).

Metadata per column (reference metadata/metadata.py): {"shape": [...],
"padding": value} — shape [] scalar, [L] 1-D array, [L, W] 2-D array.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import pyarrow as pa
import torch


def mask_name(column: str) -> str:
    return f"{column}_mask"


class NumericColumn:
    def __init__(self, name: str) -> None:
        self.name = name

    def decode(self, array) -> Dict[str, torch.Tensor]:
        np_arr = array.to_numpy(zero_copy_only=False)
        tensor = torch.from_numpy(np.ascontiguousarray(np_arr))
        if tensor.dtype in (torch.int8, torch.int16, torch.int32, torch.uint8):
            tensor = tensor.long()
        return {self.name: tensor}


class Array1DColumn:
    """Ragged list column -> [B, L] right-padded + bool validity mask."""

    def __init__(self, name: str, length: int, padding=0) -> None:
        self.name = name
        self.length = length
        self.padding = padding

    def decode(self, array) -> Dict[str, torch.Tensor]:
        if isinstance(array, pa.ChunkedArray):
            array = array.combine_chunks()
        values = array.values.to_numpy(zero_copy_only=False)
        offsets = array.offsets.to_numpy(zero_copy_only=False).astype(np.int64)
        B, L = len(array), self.length
        lengths = np.minimum(offsets[1:] - offsets[:-1], L)
        dtype = values.dtype if values.dtype.kind == "f" else np.int64
        out = np.full((B, L), self.padding, dtype=dtype)
        mask = np.zeros((B, L), dtype=bool)
        # truncate KEEPING THE TAIL (most recent interactions), like the
        # reference window slicing (torch_sequential_dataset.py:115)
        for i in range(B):
            n = lengths[i]
            start = offsets[i + 1] - n
            out[i, :n] = values[start : offsets[i + 1]]
            mask[i, :n] = True
        return {self.name: torch.from_numpy(out), mask_name(self.name): torch.from_numpy(mask)}


class Array2DColumn:
    """Nested list column -> [B, L, W] + [B, L] mask (reference :22)."""

    def __init__(self, name: str, length: int, width: int, padding=0) -> None:
        self.name = name
        self.length = length
        self.width = width
        self.padding = padding

    def decode(self, array) -> Dict[str, torch.Tensor]:
        if isinstance(array, pa.ChunkedArray):
            array = array.combine_chunks()
        B, L, W = len(array), self.length, self.width
        outer_offsets = array.offsets.to_numpy(zero_copy_only=False).astype(np.int64)
        inner = array.values  # list array
        inner_offsets = inner.offsets.to_numpy(zero_copy_only=False).astype(np.int64)
        values = inner.values.to_numpy(zero_copy_only=False)
        dtype = values.dtype if values.dtype.kind == "f" else np.int64
        out = np.full((B, L, W), self.padding, dtype=dtype)
        mask = np.zeros((B, L), dtype=bool)
        for i in range(B):
            n = min(outer_offsets[i + 1] - outer_offsets[i], L)
            row_start = outer_offsets[i + 1] - n
            for j in range(n):
                a = inner_offsets[row_start + j]
                b = min(inner_offsets[row_start + j + 1], a + W)
                out[i, j, : b - a] = values[a:b]
                mask[i, j] = True
        return {self.name: torch.from_numpy(out), mask_name(self.name): torch.from_numpy(mask)}


class NamedColumns:
    """Column decoders for a table, built from per-column metadata."""

    def __init__(self, metadata: Dict[str, Dict], columns: Optional[list] = None) -> None:
        self.metadata = metadata
        self.columns = columns

    def decoder_for(self, name: str):
        meta = self.metadata.get(name, {})
        shape = meta.get("shape", [])
        padding = meta.get("padding", 0)
        if len(shape) == 0:
            return NumericColumn(name)
        if len(shape) == 1:
            return Array1DColumn(name, shape[0], padding)
        if len(shape) == 2:
            return Array2DColumn(name, shape[0], shape[1], padding)
        raise ValueError(f"Unsupported shape {shape} for column {name}")

    def decode_table(self, table: pa.Table) -> Dict[str, torch.Tensor]:
        out: Dict[str, torch.Tensor] = {}
        names = self.columns or table.column_names
        for name in names:
            if name not in table.column_names:
                continue
            out.update(self.decoder_for(name).decode(table.column(name)))
        return out


def collate_batches(batches):
    """Concatenate a list of dict-of-tensor batches (reference collate.py)."""
    if len(batches) == 1:
        return batches[0]
    keys = batches[0].keys()
    return {k: torch.cat([b[k] for b in batches]) for k in keys}
