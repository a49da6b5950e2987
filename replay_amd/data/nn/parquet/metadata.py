"""Parquet column-metadata helpers (reference
replay/data/nn/parquet/metadata/metadata.py): a column's metadata dict says
whether it decodes as a scalar, a [L] list or a [L, W] list-of-lists and
what padding value fills the ragged tail.

Format (shared with NamedColumns, columns.py): ``{"shape": [...],
"padding": value}`` — shape ``[]``/absent = scalar, ``[L]`` or bare int =
1-D array, ``[L, W]`` = 2-D array.
"""

from __future__ import annotations

from typing import Any, Dict, List

ColumnMetadata = Dict[str, Any]
Metadata = Dict[str, ColumnMetadata]

SHAPE_FLAG = "shape"
PADDING_FLAG = "padding"
DEFAULT_PADDING = 0


def _shape_of(column_metadata: ColumnMetadata) -> List[int]:
    value = column_metadata.get(SHAPE_FLAG, [])
    if isinstance(value, int):
        return [value]
    return list(value)


def is_array_1d(column_metadata: ColumnMetadata) -> bool:
    shape = _shape_of(column_metadata)
    return len(shape) == 1 and all(isinstance(v, int) for v in shape)


def is_array_2d(column_metadata: ColumnMetadata) -> bool:
    shape = _shape_of(column_metadata)
    return len(shape) == 2 and all(isinstance(v, int) for v in shape)


def is_number(column_metadata: ColumnMetadata) -> bool:
    return not is_array_1d(column_metadata) and not is_array_2d(column_metadata)


def _listing(metadata: Metadata, check) -> List[str]:
    return sorted(name for name, meta in metadata.items() if check(meta))


def get_numeric_columns(metadata: Metadata) -> List[str]:
    return _listing(metadata, is_number)


def get_1d_array_columns(metadata: Metadata) -> List[str]:
    return _listing(metadata, is_array_1d)


def get_2d_array_columns(metadata: Metadata) -> List[str]:
    return _listing(metadata, is_array_2d)


def get_padding(metadata: Metadata, column_name: str) -> Any:
    if column_name not in metadata:
        raise KeyError(f"Column {column_name} not found in metadata.")
    return metadata[column_name].get(PADDING_FLAG, DEFAULT_PADDING)


def get_shape(metadata: Metadata, column_name: str):
    if column_name not in metadata:
        raise KeyError(f"Column {column_name} not found in metadata.")
    if is_number(metadata[column_name]):
        raise ValueError(f"Column {column_name} is not an array.")
    result = metadata[column_name][SHAPE_FLAG]
    for i, v in enumerate(result if isinstance(result, list) else [result]):
        if v < 1:
            raise ValueError(
                f"Shape for column {column_name} at position {i} is not a positive integer."
            )
    return result
