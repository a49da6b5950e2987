"""Off-policy evaluation utilities.

Parity target: the reference wraps the Open Bandit Pipeline
(replay/experimental/scenarios/obp_wrapper/replay_offline.py, 272 LoC).  OBP
is not installable offline; the same use case (evaluate a new policy on
logged bandit feedback) is served by native IPS / SNIPS estimators and an
``OBPOfflinePolicyLearner``-shaped trainer over logged (context, action,
reward, propensity) tuples.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

OBP_AVAILABLE = False  # the obp library itself is not in the offline stack


def ips_estimate(
    logged: pd.DataFrame,
    policy_prob: np.ndarray,
    reward_column: str = "reward",
    propensity_column: str = "propensity",
) -> float:
    """Inverse-propensity-score value estimate of a policy:
    V = mean(r * pi(a|x) / mu(a|x))."""
    w = policy_prob / np.maximum(logged[propensity_column].to_numpy(), 1e-12)
    return float((logged[reward_column].to_numpy() * w).mean())


def snips_estimate(
    logged: pd.DataFrame,
    policy_prob: np.ndarray,
    reward_column: str = "reward",
    propensity_column: str = "propensity",
) -> float:
    """Self-normalized IPS (lower variance)."""
    w = policy_prob / np.maximum(logged[propensity_column].to_numpy(), 1e-12)
    denom = w.sum()
    if denom == 0:
        return 0.0
    return float((logged[reward_column].to_numpy() * w).sum() / denom)


class OBPOfflinePolicyLearner:
    """Learns a softmax policy over items from logged bandit feedback via an
    IPS-weighted classifier; evaluates with ips/snips."""

    def __init__(self, n_actions: int, len_list: int = 1, seed: Optional[int] = None) -> None:
        self.n_actions = n_actions
        self.len_list = len_list
        self.seed = seed
        self._clf = None

    def fit(
        self,
        context: np.ndarray,  # [N, d]
        action: np.ndarray,  # [N]
        reward: np.ndarray,  # [N]
        pscore: Optional[np.ndarray] = None,
    ) -> "OBPOfflinePolicyLearner":
        from sklearn.linear_model import LogisticRegression

        pscore = pscore if pscore is not None else np.full(len(action), 1.0 / self.n_actions)
        weights = reward / np.maximum(pscore, 1e-12)
        mask = weights > 0
        if mask.sum() < 2 or len(np.unique(action[mask])) < 2:
            self._clf = None
            return self
        self._clf = LogisticRegression(max_iter=200, random_state=self.seed)
        self._clf.fit(context[mask], action[mask], sample_weight=weights[mask])
        return self

    def predict(self, context: np.ndarray) -> np.ndarray:
        """Action distribution [N, n_actions, len_list]."""
        n = len(context)
        if self._clf is None:
            probs = np.full((n, self.n_actions), 1.0 / self.n_actions)
        else:
            probs = np.zeros((n, self.n_actions))
            probs[:, self._clf.classes_] = self._clf.predict_proba(context)
        return probs[:, :, None].repeat(self.len_list, axis=2)
