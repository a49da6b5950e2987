"""Numerical-feature discretization.

Parity with reference replay/preprocessing/discretizer.py:
GreedyDiscretizingRule:63, QuantileDiscretizingRule:376, Discretizer:603.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import numpy as np
import pandas as pd


HandleInvalidStrategies = ("error", "skip", "keep")  # reference discretizer.py:25


class BaseDiscretizingRule:
    is_fitted = False

    def fit(self, df: pd.DataFrame) -> "BaseDiscretizingRule":  # pragma: no cover
        raise NotImplementedError

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:  # pragma: no cover
        raise NotImplementedError


class QuantileDiscretizingRule(BaseDiscretizingRule):
    """Quantile bucketing into n_bins (reference discretizer.py:376)."""

    def __init__(self, column: str, n_bins: int = 10, handle_invalid: str = "error") -> None:
        if handle_invalid not in ("error", "skip", "keep"):
            raise ValueError("handle_invalid must be error/skip/keep")
        self.column = column
        self.n_bins = n_bins
        self.handle_invalid = handle_invalid
        self._bins: Optional[np.ndarray] = None

    def fit(self, df: pd.DataFrame) -> "QuantileDiscretizingRule":
        values = df[self.column].dropna().to_numpy(dtype=np.float64)
        quantiles = np.quantile(values, np.linspace(0, 1, self.n_bins + 1))
        self._bins = np.unique(quantiles)
        self.is_fitted = True
        return self

    def _assign(self, values: np.ndarray) -> np.ndarray:
        inner = self._bins[1:-1]
        return np.searchsorted(inner, values, side="right").astype(np.int64)

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if not self.is_fitted:
            raise RuntimeError("Rule is not fitted")
        out = df.copy()
        col = out[self.column]
        nan_mask = col.isna()
        if nan_mask.any():
            if self.handle_invalid == "error":
                raise ValueError(f"NaN values in column {self.column}")
            if self.handle_invalid == "skip":
                out = out[~nan_mask]
                col = out[self.column]
                nan_mask = col.isna()
        codes = self._assign(col.fillna(0).to_numpy(dtype=np.float64))
        if nan_mask.any():  # keep: NaN bucket = n_bins
            codes[nan_mask.to_numpy()] = len(self._bins) - 1
        out[self.column] = codes
        return out


class GreedyDiscretizingRule(BaseDiscretizingRule):
    """Greedy equal-mass bucketing that respects heavy repeated values
    (reference discretizer.py:63).  Builds buckets left-to-right over the
    value histogram so each bucket gets ~len/n_bins points without splitting
    a single repeated value across buckets."""

    def __init__(self, column: str, n_bins: int = 10, min_data_in_bin: int = 1, handle_invalid: str = "error") -> None:
        if handle_invalid not in ("error", "skip", "keep"):
            raise ValueError("handle_invalid must be error/skip/keep")
        self.column = column
        self.n_bins = n_bins
        self.min_data_in_bin = min_data_in_bin
        self.handle_invalid = handle_invalid
        self._upper_bounds: Optional[np.ndarray] = None

    def fit(self, df: pd.DataFrame) -> "GreedyDiscretizingRule":
        values = df[self.column].dropna().to_numpy(dtype=np.float64)
        uniq, counts = np.unique(values, return_counts=True)
        total = counts.sum()
        target = max(total / self.n_bins, self.min_data_in_bin)
        bounds: List[float] = []
        acc = 0
        for v, c in zip(uniq, counts):
            acc += c
            if acc >= target and len(bounds) < self.n_bins - 1:
                bounds.append(float(v))
                acc = 0
        self._upper_bounds = np.asarray(bounds)
        self.is_fitted = True
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        if not self.is_fitted:
            raise RuntimeError("Rule is not fitted")
        out = df.copy()
        col = out[self.column]
        nan_mask = col.isna()
        if nan_mask.any():
            if self.handle_invalid == "error":
                raise ValueError(f"NaN values in column {self.column}")
            if self.handle_invalid == "skip":
                out = out[~nan_mask]
                col = out[self.column]
                nan_mask = col.isna()
        codes = np.searchsorted(self._upper_bounds, col.fillna(0).to_numpy(dtype=np.float64), side="right").astype(
            np.int64
        )
        if nan_mask.any():
            codes[nan_mask.to_numpy()] = len(self._upper_bounds) + 1
        out[self.column] = codes
        return out


class Discretizer:
    """Multi-column discretizer composed of rules (reference discretizer.py:603)."""

    def __init__(self, rules: Sequence[BaseDiscretizingRule]) -> None:
        self.rules = list(rules)

    def fit(self, df: pd.DataFrame) -> "Discretizer":
        for rule in self.rules:
            rule.fit(df)
        return self

    def transform(self, df: pd.DataFrame) -> pd.DataFrame:
        for rule in self.rules:
            df = rule.transform(df)
        return df

    def fit_transform(self, df: pd.DataFrame) -> pd.DataFrame:
        return self.fit(df).transform(df)
