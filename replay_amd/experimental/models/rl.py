"""Offline-RL recommenders.

Parity targets: CQL (reference experimental/models/cql.py, 454 LoC — the
reference delegates to d3rlpy, unavailable offline) and DDPG
(experimental/models/ddpg.py, 932 LoC).  Native single-step (contextual)
implementations: state = aggregate of the user's interacted-item embeddings;
CQL learns a discrete conservative Q over items, DDPG learns an actor in
item-embedding space with a critic, recommending items nearest to the
actor's output.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from replay_amd.models.base_rec import Recommender


class _StateBuilder:
    """state[u] = mean embedding of the user's items (built after fit)."""

    @staticmethod
    def build(inter, q_col, i_col, item_emb, n_users):
        import torch

        E = item_emb.shape[1]
        states = torch.zeros(n_users, E)
        counts = torch.zeros(n_users, 1)
        u = torch.from_numpy(inter[q_col].to_numpy(dtype=np.int64))
        i = torch.from_numpy(inter[i_col].to_numpy(dtype=np.int64))
        states.index_add_(0, u, item_emb[i])
        counts.index_add_(0, u, torch.ones(len(u), 1))
        return states / counts.clamp(min=1)


class CQL(Recommender):
    """Discrete conservative Q-learning (Kumar et al. 2020):
    loss = MSE(Q(s,a), r) + alpha * (logsumexp_a' Q(s,a') - Q(s,a))."""

    def __init__(
        self,
        embedding_dim: int = 32,
        hidden_dim: int = 64,
        alpha: float = 1.0,
        epochs: int = 5,
        learning_rate: float = 1e-3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
        n_epochs: Optional[int] = None,  # reference d3rlpy arg name
    ) -> None:
        super().__init__()
        self.embedding_dim = embedding_dim
        self.hidden_dim = hidden_dim
        self.alpha = alpha
        self.epochs = n_epochs or epochs
        self.learning_rate = learning_rate
        self.seed = seed
        self.device_arg = device

    @property
    def _init_args(self):
        return {
            "embedding_dim": self.embedding_dim,
            "hidden_dim": self.hidden_dim,
            "alpha": self.alpha,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "seed": self.seed,
        }

    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions
        n_i, E, H = self._item_dim_size, self.embedding_dim, self.hidden_dim
        item_emb = torch.nn.Embedding(n_i, E)
        state_proj = torch.nn.Sequential(torch.nn.Linear(E, H), torch.nn.ReLU())
        q_head = torch.nn.Linear(H, n_i)  # discrete Q over all items
        net = torch.nn.ModuleList([item_emb, state_proj, q_head]).to(device)
        opt = torch.optim.Adam(net.parameters(), lr=self.learning_rate)

        users = torch.from_numpy(inter[self.query_column].to_numpy(dtype=np.int64))
        items = torch.from_numpy(inter[self.item_column].to_numpy(dtype=np.int64))
        rewards = (
            torch.from_numpy(inter[self.rating_column].to_numpy(dtype=np.float32))
            if self.rating_column in inter.columns
            else torch.ones(len(inter))
        )
        for _ in range(self.epochs):
            states = _StateBuilder.build(
                inter, self.query_column, self.item_column, item_emb.weight.detach().cpu(), self._query_dim_size
            ).to(device)
            perm = torch.randperm(len(users))
            for s in range(0, len(perm), 4096):
                b = perm[s : s + 4096]
                st = state_proj(states[users[b].to(device)])
                q_all = q_head(st)  # [B, n_items]
                q_sa = q_all.gather(1, items[b].to(device).unsqueeze(1)).squeeze(1)
                td = torch.nn.functional.mse_loss(q_sa, rewards[b].to(device))
                conservative = (torch.logsumexp(q_all, dim=1) - q_sa).mean()
                loss = td + self.alpha * conservative
                opt.zero_grad()
                loss.backward()
                opt.step()
        self._item_emb = item_emb.weight.detach().cpu()
        self._state_proj = state_proj.cpu().eval()
        self._q_head = q_head.cpu().eval()
        self._inter = inter

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        states = _StateBuilder.build(
            dataset.interactions, self.query_column, self.item_column, self._item_emb, self._query_dim_size
        )
        with torch.no_grad():
            q_all = self._q_head(self._state_proj(states[q_ids]))
        scores = q_all.numpy()[:, i_ids]
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))


class DDPG(Recommender):
    """Deterministic actor-critic: actor maps the user state to a vector in
    item-embedding space; recommendations = nearest item embeddings."""

    def __init__(
        self,
        embedding_dim: int = 32,
        hidden_dim: int = 64,
        epochs: int = 5,
        learning_rate: float = 1e-3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
        noise_sigma: float = 0.1,
    ) -> None:
        super().__init__()
        self.embedding_dim = embedding_dim
        self.hidden_dim = hidden_dim
        self.epochs = epochs
        self.learning_rate = learning_rate
        self.seed = seed
        self.device_arg = device
        self.noise_sigma = noise_sigma

    @property
    def _init_args(self):
        return {
            "embedding_dim": self.embedding_dim,
            "hidden_dim": self.hidden_dim,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "seed": self.seed,
            "noise_sigma": self.noise_sigma,
        }

    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions
        n_i, E, H = self._item_dim_size, self.embedding_dim, self.hidden_dim
        item_emb = torch.nn.Embedding(n_i, E)
        actor = torch.nn.Sequential(torch.nn.Linear(E, H), torch.nn.ReLU(), torch.nn.Linear(H, E))
        critic = torch.nn.Sequential(torch.nn.Linear(2 * E, H), torch.nn.ReLU(), torch.nn.Linear(H, 1))
        nets = torch.nn.ModuleList([item_emb, actor, critic]).to(device)
        opt_c = torch.optim.Adam(list(critic.parameters()) + list(item_emb.parameters()), lr=self.learning_rate)
        opt_a = torch.optim.Adam(actor.parameters(), lr=self.learning_rate)

        users = torch.from_numpy(inter[self.query_column].to_numpy(dtype=np.int64))
        items = torch.from_numpy(inter[self.item_column].to_numpy(dtype=np.int64))
        rewards = (
            torch.from_numpy(inter[self.rating_column].to_numpy(dtype=np.float32))
            if self.rating_column in inter.columns
            else torch.ones(len(inter))
        )
        for _ in range(self.epochs):
            states = _StateBuilder.build(
                inter, self.query_column, self.item_column, item_emb.weight.detach().cpu(), self._query_dim_size
            ).to(device)
            perm = torch.randperm(len(users))
            for s in range(0, len(perm), 4096):
                b = perm[s : s + 4096]
                st = states[users[b].to(device)]
                act = item_emb(items[b].to(device))
                # critic: fit reward of taken action
                q = critic(torch.cat([st, act], dim=-1)).squeeze(-1)
                c_loss = torch.nn.functional.mse_loss(q, rewards[b].to(device))
                opt_c.zero_grad()
                c_loss.backward()
                opt_c.step()
                # actor: maximize critic value of proposed action
                a_loss = -critic(torch.cat([st, actor(st)], dim=-1)).mean()
                opt_a.zero_grad()
                a_loss.backward()
                opt_a.step()
        self._item_emb = item_emb.weight.detach().cpu()
        self._actor = actor.cpu().eval()

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        states = _StateBuilder.build(
            dataset.interactions, self.query_column, self.item_column, self._item_emb, self._query_dim_size
        )
        with torch.no_grad():
            actions = self._actor(states[q_ids])  # [Q, E]
        scores = (actions @ self._item_emb[i_ids].T).numpy()
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))
