"""Label encoding for id and categorical columns.

Parity with the reference encoder (replay/preprocessing/label_encoder.py:
LabelEncodingRule:86, SequenceEncodingRule:568, LabelEncoder:794): per-column
rules, ``partial_fit``, unknown-value strategies ``error`` /
``use_default_value`` / ``drop``, ``inverse_transform`` and save/load.
Implementation is pandas/numpy-native (no Spark tier).
"""

from __future__ import annotations

import json
import warnings
from pathlib import Path
from typing import Dict, List, Optional, Sequence, Union

import numpy as np
import pandas as pd

HandleUnknownStrategies = ("error", "use_default_value", "drop")


class LabelEncoderPartialFitWarning(Warning):
    """partial_fit saw already-known labels (reference label_encoder.py)."""


class LabelEncoderTransformWarning(Warning):
    """Warning raised on unseen labels with non-error strategies."""


class LabelEncodingRule:
    """Encode one scalar column to contiguous ids [0..n)."""

    is_sequence = False

    def __init__(
        self,
        column: str,
        mapping: Optional[Dict] = None,
        handle_unknown: str = "error",
        default_value: Optional[Union[int, str]] = None,
    ) -> None:
        if handle_unknown not in HandleUnknownStrategies:
            raise ValueError(f"handle_unknown must be one of {HandleUnknownStrategies}")
        if handle_unknown == "use_default_value" and default_value is not None:
            if not (default_value == "last" or isinstance(default_value, int)):
                raise ValueError("default_value must be int, 'last' or None")
        self._col = column
        self._handle_unknown = handle_unknown
        self._default_value = default_value
        self._mapping: Optional[Dict] = dict(mapping) if mapping is not None else None
        self._inverse: Optional[List] = None
        if self._mapping is not None:
            self._rebuild_inverse()

    # -- properties ------------------------------------------------------------
    @property
    def column(self) -> str:
        return self._col

    @property
    def mapping(self) -> Optional[Dict]:
        return self._mapping

    def get_default_value(self) -> Optional[int]:
        if self._default_value == "last":
            return len(self._mapping) if self._mapping is not None else None
        return self._default_value

    def _rebuild_inverse(self) -> None:
        inverse = [None] * len(self._mapping)
        for key, code in self._mapping.items():
            inverse[code] = key
        self._inverse = inverse

    # -- fitting ---------------------------------------------------------------
    def _column_values(self, df: pd.DataFrame) -> np.ndarray:
        return df[self._col].to_numpy()

    def fit(self, df: pd.DataFrame) -> "LabelEncodingRule":
        if self._mapping is None:
            # reference label_encoder.py:188: unique values are SORTED before
            # code assignment so mappings are order-independent (mixed-type
            # columns, which plain sorting cannot order, fall back to repr)
            try:
                uniques = df[self._col].sort_values().drop_duplicates().tolist()
            except TypeError:
                uniques = sorted(set(df[self._col].tolist()), key=repr)
            self._mapping = {value: idx for idx, value in enumerate(uniques)}
            self._rebuild_inverse()
        return self

    def partial_fit(self, df: pd.DataFrame) -> "LabelEncodingRule":
        if self._mapping is None:
            return self.fit(df)
        new_values = set(df[self._col].tolist()) - set(self._mapping)
        if not new_values:
            warnings.warn(
                "partial_fit will have no effect because there are no new "
                f"values in the incoming dataset at '{self.column}' column",
                LabelEncoderPartialFitWarning,
            )
            return self
        next_code = len(self._mapping)
        for value in sorted(new_values, key=repr):
            self._mapping[value] = next_code
            next_code += 1
        self._rebuild_inverse()
        return self

    # -- transforms ------------------------------------------------------------
    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if self._mapping is None:
            raise RuntimeError(f"Rule for column {self._col} is not fitted")
        col = df[self._col]
        encoded = col.map(self._mapping)
        mask_unknown = encoded.isna() & col.notna()
        if mask_unknown.any():
            unknown = col[mask_unknown].unique().tolist()
            if self._handle_unknown == "error":
                raise ValueError(f"Unseen labels in column {self._col}: {unknown[:10]}")
            if self._handle_unknown == "drop":
                import warnings

                warnings.warn(
                    f"Dropping {int(mask_unknown.sum())} rows with unseen labels in {self._col}",
                    LabelEncoderTransformWarning,
                )
                df = df[~mask_unknown.to_numpy()]
                encoded = encoded[~mask_unknown.to_numpy()]
            else:  # use_default_value
                import warnings

                warnings.warn(
                    f"Replacing unseen labels in {self._col} with default value",
                    LabelEncoderTransformWarning,
                )
                encoded = encoded.fillna(self.get_default_value())
        out = df.copy()
        out[self._col] = encoded.to_numpy()
        if len(out):
            try:
                out[self._col] = out[self._col].astype(np.int64)
            except (TypeError, ValueError):
                pass
        return out

    def fit_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        return self.fit(df).transform(df)

    def inverse_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if self._inverse is None:
            raise RuntimeError(f"Rule for column {self._col} is not fitted")
        out = df.copy()
        inverse = self._inverse
        n = len(inverse)
        out[self._col] = [inverse[int(v)] if 0 <= int(v) < n else None for v in out[self._col]]
        return out

    def set_handle_unknown(self, handle_unknown: str) -> None:
        if handle_unknown not in HandleUnknownStrategies:
            raise ValueError(f"handle_unknown must be one of {HandleUnknownStrategies}")
        self._handle_unknown = handle_unknown

    def set_default_value(self, default_value: Optional[Union[int, str]]) -> None:
        self._default_value = default_value

    # -- serialization ---------------------------------------------------------
    def _state(self) -> Dict:
        keys = list(self._mapping.keys()) if self._mapping is not None else None
        return {
            "rule_type": type(self).__name__,
            "column": self._col,
            "handle_unknown": self._handle_unknown,
            "default_value": self._default_value,
            "mapping_keys": keys,
        }

    @staticmethod
    def _from_state(state: Dict) -> "LabelEncodingRule":
        cls = {"LabelEncodingRule": LabelEncodingRule, "SequenceEncodingRule": SequenceEncodingRule}[
            state["rule_type"]
        ]
        mapping = None
        if state["mapping_keys"] is not None:
            mapping = {key: idx for idx, key in enumerate(state["mapping_keys"])}
        return cls(
            column=state["column"],
            mapping=mapping,
            handle_unknown=state["handle_unknown"],
            default_value=state["default_value"],
        )


class SequenceEncodingRule(LabelEncodingRule):
    """Encode a column whose cells are lists of labels.

    Reference: replay/preprocessing/label_encoder.py:568.
    """

    is_sequence = True

    def fit(self, df: pd.DataFrame) -> "SequenceEncodingRule":
        if self._mapping is None:
            # reference convention: exploded unique values sorted before
            # code assignment (label_encoder.py:188 via fit on the explode)
            values = {value for seq in df[self._col] for value in seq}
            try:
                uniques = sorted(values)
            except TypeError:
                uniques = sorted(values, key=repr)
            self._mapping = {value: idx for idx, value in enumerate(uniques)}
            self._rebuild_inverse()
        return self

    def partial_fit(self, df: pd.DataFrame) -> "SequenceEncodingRule":
        if self._mapping is None:
            return self.fit(df)
        code = len(self._mapping)
        for seq in df[self._col]:
            for value in seq:
                if value not in self._mapping:
                    self._mapping[value] = code
                    code += 1
        self._rebuild_inverse()
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if self._mapping is None:
            raise RuntimeError(f"Rule for column {self._col} is not fitted")
        mapping = self._mapping
        default = self.get_default_value()
        handle = self._handle_unknown
        has_unknown = False

        def encode(seq):
            nonlocal has_unknown
            out = []
            for value in seq:
                code = mapping.get(value)
                if code is None:
                    has_unknown = True
                    if handle == "error":
                        raise ValueError(f"Unseen label {value!r} in sequence column {self._col}")
                    if handle == "drop":
                        continue
                    code = default
                out.append(code)
            return np.asarray(out, dtype=np.int64)

        out = df.copy()
        out[self._col] = [encode(seq) for seq in df[self._col]]
        if has_unknown and handle != "error":
            import warnings

            warnings.warn(f"Unseen labels in sequence column {self._col}", LabelEncoderTransformWarning)
        return out

    def inverse_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if self._inverse is None:
            raise RuntimeError(f"Rule for column {self._col} is not fitted")
        inverse = self._inverse
        n = len(inverse)
        out = df.copy()
        out[self._col] = [[inverse[int(v)] for v in seq if 0 <= int(v) < n] for seq in df[self._col]]
        return out


class LabelEncoder:
    """Multi-column encoder composed of rules (reference label_encoder.py:794)."""

    def __init__(self, rules: Sequence[LabelEncodingRule]) -> None:
        self.rules = list(rules)

    @property
    def mapping(self) -> Dict[str, Dict]:
        return {rule.column: rule.mapping for rule in self.rules}

    @property
    def inverse_mapping(self) -> Dict[str, Dict]:
        return {
            rule.column: {code: key for key, code in rule.mapping.items()} for rule in self.rules if rule.mapping
        }

    def fit(self, df: pd.DataFrame) -> "LabelEncoder":
        for rule in self.rules:
            rule.fit(df)
        return self

    def partial_fit(self, df: pd.DataFrame) -> "LabelEncoder":
        for rule in self.rules:
            rule.partial_fit(df)
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        for rule in self.rules:
            df = rule.transform(df)
        return df

    def fit_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        return self.fit(df).transform(df)

    def inverse_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        for rule in self.rules:
            df = rule.inverse_transform(df)
        return df

    def set_handle_unknowns(self, handle_unknown: Dict[str, str]) -> None:
        by_col = {rule.column: rule for rule in self.rules}
        for column, strategy in handle_unknown.items():
            if column not in by_col:
                raise ValueError(f"No rule for column {column}")
            by_col[column].set_handle_unknown(strategy)

    def set_default_values(self, default_values: Dict[str, Union[int, str, None]]) -> None:
        by_col = {rule.column: rule for rule in self.rules}
        for column, value in default_values.items():
            if column not in by_col:
                raise ValueError(f"No rule for column {column}")
            by_col[column].set_default_value(value)

    # -- serialization ---------------------------------------------------------
    def save(self, path: Union[str, Path]) -> None:
        import pickle

        base = Path(path)
        base.mkdir(parents=True, exist_ok=True)
        state = {"rules": [rule._state() for rule in self.rules]}
        with open(base / "label_encoder.pkl", "wb") as f:
            pickle.dump(state, f)

    @classmethod
    def load(cls, path: Union[str, Path]) -> "LabelEncoder":
        import pickle

        base = Path(path)
        with open(base / "label_encoder.pkl", "rb") as f:
            state = pickle.load(f)
        return cls([LabelEncodingRule._from_state(s) for s in state["rules"]])
