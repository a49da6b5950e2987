"""Array-column padding for pandas frames (behavioral parity with the
reference's experimental Padder, experimental/preprocessing/padder.py:11;
the Spark branch is out of scope per SURVEY §7)."""

from __future__ import annotations

from typing import Iterable, List, Optional, Union

import pandas as pd


class Padder:
    """Pad (and optionally cut) list columns to a fixed size.

    ``padding_side`` picks where the fill goes; ``cut_side="right"`` keeps
    the TAIL of an over-long list (the reference's convention for recency),
    ``"left"`` keeps the head.
    """

    def __init__(
        self,
        pad_columns: Union[str, List[str]],
        padding_side: str = "right",
        padding_value: Union[str, float, list, None] = 0,
        array_size: Optional[int] = None,
        cut_array: bool = True,
        cut_side: str = "right",
    ) -> None:
        if padding_side not in ("right", "left"):
            raise ValueError("padding_side should be 'right' or 'left'")
        if cut_side not in ("right", "left"):
            raise ValueError("cut_side should be 'right' or 'left'")
        self.pad_columns = [pad_columns] if isinstance(pad_columns, str) else list(pad_columns)
        self.padding_side = padding_side
        values = (
            list(padding_value)
            if isinstance(padding_value, Iterable) and not isinstance(padding_value, str)
            else [padding_value]
        )
        if len(values) == 1 and len(self.pad_columns) > 1:
            values = values * len(self.pad_columns)
        if len(values) != len(self.pad_columns):
            raise ValueError("pad_columns and padding_value should have same length")
        self.padding_value = values
        if array_size is not None and (not isinstance(array_size, int) or array_size < 1):
            raise ValueError("array_size should be positive integer greater than 0")
        self.array_size = array_size
        self.cut_array = cut_array
        self.cut_side = cut_side

    def transform(self, interactions: pd.DataFrame) -> pd.DataFrame:
        res = interactions.copy(deep=True)
        for col, pad_value in zip(self.pad_columns, self.padding_value):
            if col not in res.columns:
                raise ValueError(f"Column {col} not in DataFrame columns.")
            lists = res[col].apply(lambda x: x if isinstance(x, list) else [])
            size = self.array_size or int(lists.str.len().max() or 0)
            if self.cut_array:
                if self.cut_side == "right":
                    lists = lists.apply(lambda x: x[-min(len(x), size):])
                else:
                    lists = lists.apply(lambda x: x[: min(len(x), size)])
            pads = lists.apply(lambda x: [pad_value] * max(0, size - len(x)))
            res[col] = lists + pads if self.padding_side == "right" else pads + lists
        return res
