"""Legacy metric base classes (reference experimental/metrics/
base_metric.py:193 ``Metric.__call__(recs, ground_truth, k)`` and :441
``NCISMetric`` counterfactual weighting).  Pandas-native: the reference's
Spark-UDF/Scala paths (base_metric.py:385) are replaced by vectorized
pandas; the per-user metric math delegates to ``replay_amd.metrics``."""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Dict, Iterable, Optional, Union

import numpy as np
import pandas as pd

IntOrList = Union[Iterable[int], int]

DEFAULT_COLUMNS = {"query_column": "user_idx", "item_column": "item_idx", "rating_column": "relevance"}


class Metric(ABC):
    """Legacy callable metric over [user_idx, item_idx, relevance] frames."""

    _main_metric_name: Optional[str] = None  # name in replay_amd.metrics

    def __init__(self, use_scala_udf: bool = False) -> None:
        # JVM UDFs do not exist in the MI355X build; the flag is accepted for
        # constructor parity and ignored (pandas is the only backend)
        self._use_scala_udf = use_scala_udf

    def _mean_over_users(self, per_user: pd.Series) -> float:
        return float(per_user.mean()) if len(per_user) else 0.0

    def __call__(
        self,
        recommendations: pd.DataFrame,
        ground_truth: pd.DataFrame,
        k: IntOrList,
        ground_truth_users: Optional[pd.DataFrame] = None,
    ) -> Union[Dict[int, float], float]:
        from replay_amd.metrics import metrics as _m

        ks = [k] if isinstance(k, int) else sorted(k)
        cls = getattr(_m, self._main_metric_name or type(self).__name__)
        metric = cls(topk=ks, **DEFAULT_COLUMNS)
        recs = recommendations
        if ground_truth_users is not None:
            users = ground_truth_users["user_idx"].unique()
            recs = recs[recs["user_idx"].isin(users)]
            ground_truth = ground_truth[ground_truth["user_idx"].isin(users)]
        result = metric(recs, ground_truth)
        values = {kk: result[f"{cls.__name__}@{kk}"] for kk in ks}
        return values[ks[0]] if isinstance(k, int) else values


class NCISMetric(Metric):
    """Metric with NCIS weighting (reference base_metric.py:441): reward for
    a recommended pair is weighed by clip(current_score / previous_score,
    1/threshold, threshold); ``activation`` in {None, "sigmoid", "softmax"}
    is applied to scores first."""

    def __init__(
        self,
        prev_policy_weights: pd.DataFrame,  # [user_idx, item_idx, relevance]
        threshold: float = 10.0,
        activation: Optional[str] = None,
        use_scala_udf: bool = False,
    ) -> None:
        super().__init__(use_scala_udf)
        if activation not in (None, "sigmoid", "softmax"):
            raise ValueError(f"Unexpected activation function: {activation}")
        if threshold <= 0:
            raise ValueError("threshold must be positive")
        self._prev = prev_policy_weights.rename(columns={"relevance": "prev_relevance"})
        self.threshold = threshold
        self.activation = activation

    def _activate(self, scores: pd.Series, groups: pd.Series) -> pd.Series:
        if self.activation == "sigmoid":
            return 1.0 / (1.0 + np.exp(-scores))
        if self.activation == "softmax":
            df = pd.DataFrame({"s": scores, "g": groups})
            mx = df.groupby("g")["s"].transform("max")
            e = np.exp(df["s"] - mx)
            return e / e.groupby(df["g"]).transform("sum")
        return scores

    def weigh(self, recommendations: pd.DataFrame) -> pd.DataFrame:
        """Returns recommendations with an NCIS ``weight`` column."""
        recs = recommendations.merge(
            self._prev[["user_idx", "item_idx", "prev_relevance"]],
            on=["user_idx", "item_idx"],
            how="left",
        )
        recs["prev_relevance"] = recs["prev_relevance"].fillna(0.0)
        cur = self._activate(recs["relevance"], recs["user_idx"])
        prev = self._activate(recs["prev_relevance"], recs["user_idx"])
        with np.errstate(divide="ignore", invalid="ignore"):
            w = np.where(prev > 0, cur / prev, self.threshold)
        recs["weight"] = np.clip(w, 1.0 / self.threshold, self.threshold)
        return recs

    @abstractmethod
    def _weighted_user_metric(self, hits: np.ndarray, weights: np.ndarray, k: int) -> float:
        ...

    def __call__(
        self,
        recommendations: pd.DataFrame,
        ground_truth: pd.DataFrame,
        k: IntOrList,
        ground_truth_users: Optional[pd.DataFrame] = None,
    ) -> Union[Dict[int, float], float]:
        ks = [k] if isinstance(k, int) else sorted(k)
        recs = self.weigh(recommendations)
        if ground_truth_users is not None:
            users = ground_truth_users["user_idx"].unique()
            recs = recs[recs["user_idx"].isin(users)]
        gt_pairs = set(map(tuple, ground_truth[["user_idx", "item_idx"]].to_numpy()))
        recs = recs.sort_values(["user_idx", "relevance"], ascending=[True, False])
        values: Dict[int, float] = {}
        for kk in ks:
            per_user = []
            for _, grp in recs.groupby("user_idx", sort=False):
                top = grp.head(kk)
                hits = np.array(
                    [(u, i) in gt_pairs for u, i in top[["user_idx", "item_idx"]].to_numpy()]
                )
                per_user.append(self._weighted_user_metric(hits, top["weight"].to_numpy(), kk))
            values[kk] = float(np.mean(per_user)) if per_user else 0.0
        return values[ks[0]] if isinstance(k, int) else values
