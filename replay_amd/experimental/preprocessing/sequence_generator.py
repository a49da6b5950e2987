"""Rolling-window sequence generation for sequential models (behavioral
parity with the reference's experimental SequenceGenerator,
experimental/preprocessing/sequence_generator.py:13, pandas path): every
interaction becomes a (history window, label) training case."""

from __future__ import annotations

from typing import List, Optional, Union

import pandas as pd


class SequenceGenerator:
    """For a user history <i1, i2, i3, i4> emits cases
    (<i1> -> i2), (<i1, i2> -> i3), (<i1, i2, i3> -> i4)."""

    def __init__(
        self,
        groupby_column: Union[str, List[str]],
        orderby_column: Optional[Union[str, List[str]]] = None,
        transform_columns: Optional[Union[str, List[str]]] = None,
        len_window: int = 50,
        sequence_prefix: Optional[str] = None,
        sequence_suffix: Optional[str] = "_list",
        label_prefix: Optional[str] = "label_",
        label_suffix: Optional[str] = None,
        get_list_len: bool = False,
        list_len_column: str = "list_len",
    ) -> None:
        self.groupby_column = [groupby_column] if isinstance(groupby_column, str) else list(groupby_column)
        self.orderby_column = (
            None if orderby_column is None
            else [orderby_column] if isinstance(orderby_column, str) else list(orderby_column)
        )
        self.transform_columns = transform_columns
        self.len_window = len_window
        self.sequence_prefix = sequence_prefix or ""
        self.sequence_suffix = sequence_suffix or ""
        self.label_prefix = label_prefix or ""
        self.label_suffix = label_suffix or ""
        self.get_list_len = get_list_len
        self.list_len_column = list_len_column

    def _seq_name(self, col: str) -> str:
        return f"{self.sequence_prefix}{col}{self.sequence_suffix}"

    def _label_name(self, col: str) -> str:
        return f"{self.label_prefix}{col}{self.label_suffix}"

    def transform(self, interactions: pd.DataFrame) -> pd.DataFrame:
        cols = self.transform_columns
        if cols is None:
            cols = [c for c in interactions.columns if c not in self.groupby_column]
        elif isinstance(cols, str):
            cols = [cols]

        df = interactions.copy(deep=True)
        df.sort_values(by=self.orderby_column or self.groupby_column, inplace=True)
        for col in cols:
            # history = the up-to-len_window values BEFORE each row, per group
            seqs = []
            for _, grp in df.groupby(self.groupby_column, sort=False)[col]:
                vals = grp.tolist()
                seqs.extend(vals[max(0, i - self.len_window):i] for i in range(len(vals)))
            df[self._seq_name(col)] = seqs
            df[self._label_name(col)] = df[col]
        first_seq = self._seq_name(cols[0])
        df = df[df[first_seq].str.len() > 0]
        select = (
            self.groupby_column
            + [self._seq_name(c) for c in cols]
            + [self._label_name(c) for c in cols]
        )
        if self.get_list_len:
            df = df.copy()
            df[self.list_len_column] = df[first_seq].str.len()
            select.append(self.list_len_column)
        return df[select].reset_index(drop=True)
