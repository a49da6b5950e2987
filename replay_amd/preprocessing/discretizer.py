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
        """LightGBM-style greedy histogram binning (the reference's exact
        algorithm, discretizer.py:110-193): values whose count exceeds the
        running mean bin size become single-value bins; the rest fill bins
        to the re-estimated mean, closing early before a heavy value."""
        values = df[self.column].dropna().to_numpy(dtype=np.float64)
        uniq, counts = np.unique(values, return_counts=True)
        total = int(counts.sum())
        m = len(uniq)
        max_bin = self.n_bins + 1
        bounds: List[float] = []
        if m <= max_bin:
            acc = 0
            for i in range(m - 1):
                acc += counts[i]
                if acc >= self.min_data_in_bin:
                    bounds.append(float((uniq[i] + uniq[i + 1]) / 2.0))
                    acc = 0
        else:
            if self.min_data_in_bin > 0:
                max_bin = max(1, min(max_bin, total // self.min_data_in_bin))
            mean_size = total / max_bin
            heavy = counts >= mean_size
            light_bins = max_bin - int(heavy.sum())
            light_samples = total - int(counts[heavy].sum())
            mean_size = light_samples / light_bins
            uppers = np.full(max_bin, np.inf)
            lowers = np.full(max_bin, np.inf)
            n_closed = 0
            lowers[0] = uniq[0]
            acc = 0
            for i in range(m - 1):
                if not heavy[i]:
                    light_samples -= counts[i]
                acc += counts[i]
                close = (
                    heavy[i]
                    or acc >= mean_size
                    or (heavy[i + 1] and acc >= max(1.0, mean_size * 0.5))
                )
                if close:
                    uppers[n_closed] = uniq[i]
                    n_closed += 1
                    lowers[n_closed] = uniq[i + 1]
                    if n_closed >= max_bin - 1:
                        break
                    acc = 0
                    if not heavy[i]:
                        light_bins -= 1
                        mean_size = light_samples / light_bins
            # the last closed bin merges with the tail (reference: midpoints
            # only for the first n_closed-1 closures, then +inf)
            bounds = [float((uppers[i] + lowers[i + 1]) / 2.0) for i in range(n_closed - 1)]
        self._upper_bounds = np.asarray(bounds, dtype=np.float64)
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
