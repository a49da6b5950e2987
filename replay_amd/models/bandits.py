"""Statistical / bandit recommenders.

Parity: Wilson (reference replay/models/wilson.py), UCB (ucb.py), KLUCB
(kl_ucb.py), ThompsonSampling (thompson_sampling.py), LinUCB (lin_ucb.py:20
``DisjointArm`` per-arm A^-1/b state).

All treat ratings as binary feedback {0, 1} except where noted.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
from scipy.stats import norm

from .base_rec import HybridRecommender, NonPersonalizedRecommender


class _BinaryRatingsMixin:
    def _binary_stats(self, dataset):
        inter = dataset.interactions
        rating = inter[self.rating_column]
        vals = set(rating.unique().tolist())
        if not vals.issubset({0, 1, 0.0, 1.0}):
            raise ValueError("Ratings must be binary {0, 1}")
        grouped = inter.groupby(self.item_column)[self.rating_column].agg(["sum", "count"])
        return grouped["sum"].to_numpy(dtype=np.float64), grouped["count"].to_numpy(
            dtype=np.float64
        ), grouped.index.to_numpy()


class Wilson(_BinaryRatingsMixin, NonPersonalizedRecommender):
    """Lower bound of the Wilson confidence interval for the positive share."""

    def __init__(self, alpha: float = 0.05, add_cold_items: bool = True, cold_weight: float = 0.5) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        self.alpha = alpha

    @property
    def _init_args(self):
        return {"alpha": self.alpha, "add_cold_items": self.add_cold_items, "cold_weight": self.cold_weight}

    def _fit(self, dataset) -> None:
        pos, total, items = self._binary_stats(dataset)
        z = norm.ppf(1 - self.alpha / 2)
        p = pos / total
        denom = 1 + z**2 / total
        center = p + z**2 / (2 * total)
        margin = z * np.sqrt(p * (1 - p) / total + z**2 / (4 * total**2))
        score = (center - margin) / denom
        self.item_popularity = pd.DataFrame({self.item_column: items, self.rating_column: score})


class UCB(_BinaryRatingsMixin, NonPersonalizedRecommender):
    """UCB1: mean + coef * sqrt(2 ln N / n) (reference ucb.py)."""

    def __init__(
        self,
        exploration_coef: float = 2.0,
        sample: bool = False,
        seed: Optional[int] = None,
        add_cold_items: bool = True,
        cold_weight: float = 0.5,
    ) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        self.coef = exploration_coef
        self.sample = sample
        self.seed = seed

    @property
    def _init_args(self):
        return {
            "exploration_coef": self.coef,
            "sample": self.sample,
            "seed": self.seed,
            "add_cold_items": self.add_cold_items,
            "cold_weight": self.cold_weight,
        }

    _search_space = {"exploration_coef": {"type": "uniform", "args": [-2, 2]}}

    def _fit(self, dataset) -> None:
        pos, total, items = self._binary_stats(dataset)
        n_total = total.sum()
        score = pos / total + np.sqrt(self.coef * np.log(n_total) / total)
        self.item_popularity = pd.DataFrame({self.item_column: items, self.rating_column: score})


class KLUCB(_BinaryRatingsMixin, NonPersonalizedRecommender):
    """KL-UCB: max q such that n*KL(p, q) <= ln N + c ln ln N
    (reference kl_ucb.py), solved by bisection."""

    def __init__(
        self,
        exploration_coef: float = 0.0,
        sample: bool = False,
        seed: Optional[int] = None,
        add_cold_items: bool = True,
        cold_weight: float = 0.5,
    ) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        self.coef = exploration_coef
        self.sample = sample
        self.seed = seed

    @property
    def _init_args(self):
        return {
            "exploration_coef": self.coef,
            "sample": self.sample,
            "seed": self.seed,
            "add_cold_items": self.add_cold_items,
            "cold_weight": self.cold_weight,
        }

    @staticmethod
    def _kl(p, q):
        eps = 1e-12
        p = np.clip(p, eps, 1 - eps)
        q = np.clip(q, eps, 1 - eps)
        return p * np.log(p / q) + (1 - p) * np.log((1 - p) / (1 - q))

    def _fit(self, dataset) -> None:
        pos, total, items = self._binary_stats(dataset)
        n_total = total.sum()
        bound = (np.log(n_total) + self.coef * np.log(max(np.log(n_total), 1.0001))) / total
        p = pos / total
        lo, hi = p.copy(), np.ones_like(p)
        for _ in range(32):
            mid = (lo + hi) / 2
            mask = self._kl(p, mid) <= bound
            lo = np.where(mask, mid, lo)
            hi = np.where(mask, hi, mid)
        self.item_popularity = pd.DataFrame({self.item_column: items, self.rating_column: lo})


class ThompsonSampling(_BinaryRatingsMixin, NonPersonalizedRecommender):
    """Beta(1+pos, 1+neg) posterior sample per item (reference thompson_sampling.py)."""

    def __init__(self, sample: bool = False, seed: Optional[int] = None, add_cold_items: bool = True, cold_weight: float = 0.5) -> None:
        super().__init__(add_cold_items=add_cold_items, cold_weight=cold_weight)
        self.sample = sample
        self.seed = seed

    @property
    def _init_args(self):
        return {
            "sample": self.sample,
            "seed": self.seed,
            "add_cold_items": self.add_cold_items,
            "cold_weight": self.cold_weight,
        }

    def _fit(self, dataset) -> None:
        pos, total, items = self._binary_stats(dataset)
        rng = np.random.default_rng(self.seed)
        score = rng.beta(pos + 1, (total - pos) + 1)
        self.item_popularity = pd.DataFrame({self.item_column: items, self.rating_column: score})


class LinUCB(HybridRecommender):
    """Disjoint LinUCB over query features (reference lin_ucb.py:20).

    Per item (arm) a ridge state A = I*lambda + X^T X and b = X^T r is kept;
    score(q, a) = theta_a . x_q + alpha * sqrt(x_q^T A^-1 x_q).
    """

    def __init__(self, eps: float = 0.5, alpha: float = 1.0, regr_type: str = "disjoint", random_state: Optional[int] = None) -> None:
        super().__init__()
        if regr_type not in ("disjoint",):
            raise ValueError("only disjoint LinUCB is supported")
        self.eps = eps
        self.alpha = alpha
        self.regr_type = regr_type
        self.random_state = random_state
        self._theta: Optional[np.ndarray] = None
        self._a_inv: Optional[np.ndarray] = None
        self._arm_ids: Optional[np.ndarray] = None
        self._feature_cols = None

    @property
    def _init_args(self):
        return {"eps": self.eps, "alpha": self.alpha, "regr_type": self.regr_type, "random_state": self.random_state}

    def _query_matrix(self, dataset, query_ids: np.ndarray) -> np.ndarray:
        qf = dataset.query_features
        if qf is None:
            raise ValueError("LinUCB requires query features")
        qf = qf.set_index(self.query_column)
        cols = [c for c in qf.columns]
        self._feature_cols = cols
        return qf.loc[query_ids, cols].to_numpy(dtype=np.float64)

    def _fit(self, dataset) -> None:
        inter = dataset.interactions
        x_all = self._query_matrix(dataset, inter[self.query_column].to_numpy())
        r_all = inter[self.rating_column].to_numpy(dtype=np.float64)
        arms = inter[self.item_column].to_numpy()
        arm_ids = np.unique(arms)
        d = x_all.shape[1]
        theta = np.zeros((len(arm_ids), d))
        a_inv = np.zeros((len(arm_ids), d, d))
        for i, a in enumerate(arm_ids):
            mask = arms == a
            x = x_all[mask]
            r = r_all[mask]
            A = np.eye(d) * (1.0 + self.eps) + x.T @ x
            Ainv = np.linalg.inv(A)
            theta[i] = Ainv @ (x.T @ r)
            a_inv[i] = Ainv
        self._theta, self._a_inv, self._arm_ids = theta, a_inv, arm_ids

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy()
        x = self._query_matrix(dataset, q_ids)  # [Q, d]
        wanted = items[self.item_column].to_numpy()
        arm_positions = {a: i for i, a in enumerate(self._arm_ids)}
        sel = [arm_positions[a] for a in wanted if a in arm_positions]
        sel_ids = np.array([a for a in wanted if a in arm_positions])
        theta = self._theta[sel]  # [I, d]
        a_inv = self._a_inv[sel]  # [I, d, d]
        mean = x @ theta.T  # [Q, I]
        # exploration term: sqrt(x^T Ainv x) per (q, arm)
        xa = np.einsum("qd,ide->qie", x, a_inv)  # [Q, I, d]
        expl = np.sqrt(np.maximum(np.einsum("qie,qe->qi", xa, x), 0.0))
        scores = mean + self.alpha * expl
        return self._recs_from_scores(scores, q_ids, sel_ids, min(k, len(sel_ids)))
