"""Offline-RL recommenders.

Parity targets: CQL (reference experimental/models/cql.py — the reference
delegates to d3rlpy's SAC-based CQL over an MDPDataset built by
MdpDatasetBuilder:396-450) and DDPG (experimental/models/ddpg.py, the
reference's own 932-LoC actor-critic).  d3rlpy is unavailable offline, so
both algorithms are implemented natively with the full offline-RL
machinery the reference relies on:

* ``MdpDatasetBuilder`` — per-user EPISODES ordered by timestamp with the
  reference's reward shaping (top-K interactions per user by relevance get
  reward 1, the rest 0), terminal flags on each user's last interaction,
  and Gaussian action randomization (reference :411-447).
* CQL — double Q-heads, a target network with soft (Polyak) updates,
  discounted TD(0) targets over the episode transitions, and the
  conservative ``logsumexp(Q) - Q(s,a)`` penalty with weight ``alpha``.
* DDPG — actor + critic with target copies, soft updates, discounted TD
  targets, and Ornstein-Uhlenbeck exploration noise on the (embedding
  space) actions during fitting.
* policy save/load (reference _save_model/_load_model :350-356).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from replay_amd.models.base_rec import Recommender


class MdpDatasetBuilder:
    """User logs -> episode transitions (reference cql.py:396-450).

    Returns numpy arrays (users, items, rewards, terminals, next_index)
    where row t's next state is row t+1 of the same user episode.
    """

    def __init__(self, top_k: int = 10, action_randomization_scale: float = 1e-3) -> None:
        assert action_randomization_scale > 0
        self.top_k = top_k
        self.action_randomization_scale = action_randomization_scale

    def build(self, inter: pd.DataFrame, q_col: str, i_col: str, r_col: str, t_col: str, seed=None):
        df = inter.copy()
        if r_col not in df.columns:
            df[r_col] = 1.0
        if t_col not in df.columns:
            df[t_col] = np.arange(len(df))
        # reward: the user's top-K interactions by (relevance, timestamp)
        rank = (
            df.sort_values([r_col, t_col], ascending=[False, False], kind="stable")
            .groupby(q_col, sort=False)
            .cumcount()
        )
        df["__reward"] = (rank < self.top_k).astype(np.float32)
        df = df.sort_values([q_col, t_col], kind="stable").reset_index(drop=True)
        last_of_user = df[q_col] != df[q_col].shift(-1)
        df["__terminal"] = last_of_user.astype(np.float32)
        rng = np.random.default_rng(seed)
        actions = df[r_col].to_numpy(dtype=np.float32) + rng.normal(
            0.0, self.action_randomization_scale, len(df)
        ).astype(np.float32)
        return {
            "users": df[q_col].to_numpy(dtype=np.int64),
            "items": df[i_col].to_numpy(dtype=np.int64),
            "actions": actions,
            "rewards": df["__reward"].to_numpy(dtype=np.float32),
            "terminals": df["__terminal"].to_numpy(dtype=np.float32),
        }

    @property
    def init_args(self):
        return {"top_k": self.top_k, "action_randomization_scale": self.action_randomization_scale}


class _SeqStates:
    """Sequential episode states: state_t = mean embedding of the user's
    items BEFORE step t (zero state at episode start); next_state includes
    the current item."""

    @staticmethod
    def build(users, items, item_emb):
        import torch

        E = item_emb.shape[1]
        n = len(users)
        states = torch.zeros(n, E)
        next_states = torch.zeros(n, E)
        run = torch.zeros(E)
        count = 0
        prev_u = -1
        emb = item_emb
        for t in range(n):
            if users[t] != prev_u:
                run = torch.zeros(E)
                count = 0
                prev_u = users[t]
            states[t] = run / max(count, 1)
            run = run + emb[items[t]]
            count += 1
            next_states[t] = run / max(count, 1)
        return states, next_states

    @staticmethod
    def final_states(inter, q_col, i_col, item_emb, n_users):
        import torch

        E = item_emb.shape[1]
        states = torch.zeros(n_users, E)
        counts = torch.zeros(n_users, 1)
        u = torch.from_numpy(inter[q_col].to_numpy(dtype=np.int64))
        i = torch.from_numpy(inter[i_col].to_numpy(dtype=np.int64))
        states.index_add_(0, u, item_emb[i])
        counts.index_add_(0, u, torch.ones(len(u), 1))
        return states / counts.clamp(min=1)


def _soft_update(target, source, tau: float) -> None:
    import torch

    with torch.no_grad():
        for tp, sp in zip(target.parameters(), source.parameters()):
            tp.mul_(1.0 - tau).add_(sp, alpha=tau)


class CQL(Recommender):
    """Discrete conservative Q-learning (Kumar et al. 2020) over episode
    transitions: TD target r + gamma * max_a' Q_target(s', a') (double-Q
    min), conservative penalty alpha * (logsumexp_a Q - Q(s,a))."""

    def __init__(
        self,
        embedding_dim: int = 32,
        hidden_dim: int = 64,
        alpha: float = 1.0,
        gamma: float = 0.9,
        tau: float = 0.05,
        epochs: int = 5,
        learning_rate: float = 1e-3,
        mdp_top_k: int = 10,
        action_randomization_scale: float = 1e-3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
        n_epochs: Optional[int] = None,  # reference d3rlpy arg name
    ) -> None:
        super().__init__()
        self.embedding_dim = embedding_dim
        self.hidden_dim = hidden_dim
        self.alpha = alpha
        self.gamma = gamma
        self.tau = tau
        self.epochs = n_epochs or epochs
        self.learning_rate = learning_rate
        self.mdp_top_k = mdp_top_k
        self.action_randomization_scale = action_randomization_scale
        self.seed = seed
        self.device_arg = device

    @property
    def _init_args(self):
        return {
            "embedding_dim": self.embedding_dim,
            "hidden_dim": self.hidden_dim,
            "alpha": self.alpha,
            "gamma": self.gamma,
            "tau": self.tau,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "mdp_top_k": self.mdp_top_k,
            "action_randomization_scale": self.action_randomization_scale,
            "seed": self.seed,
        }

    def _build_nets(self, n_items):
        import torch

        E, H = self.embedding_dim, self.hidden_dim

        class QNet(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.body = torch.nn.Sequential(torch.nn.Linear(E, H), torch.nn.ReLU())
                self.q1 = torch.nn.Linear(H, n_items)
                self.q2 = torch.nn.Linear(H, n_items)

            def forward(self, s):
                h = self.body(s)
                return self.q1(h), self.q2(h)

        return QNet()

    def _fit(self, dataset) -> None:
        import copy

        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions
        n_i, E = self._item_dim_size, self.embedding_dim
        builder = MdpDatasetBuilder(self.mdp_top_k, self.action_randomization_scale)
        mdp = builder.build(
            inter, self.query_column, self.item_column, self.rating_column,
            self.timestamp_column or "timestamp", seed=self.seed,
        )
        item_emb = torch.nn.Embedding(n_i, E)
        torch.nn.init.xavier_normal_(item_emb.weight.data)
        qnet = self._build_nets(n_i).to(device)
        target = copy.deepcopy(qnet).to(device)
        for p in target.parameters():
            p.requires_grad_(False)
        opt = torch.optim.Adam(qnet.parameters(), lr=self.learning_rate)

        states, next_states = _SeqStates.build(mdp["users"], mdp["items"], item_emb.weight.detach())
        states, next_states = states.to(device), next_states.to(device)
        items = torch.from_numpy(mdp["items"]).to(device)
        rewards = torch.from_numpy(mdp["rewards"]).to(device)
        terminals = torch.from_numpy(mdp["terminals"]).to(device)
        n = len(items)
        for _ in range(self.epochs):
            perm = torch.randperm(n)
            for s in range(0, n, 4096):
                b = perm[s : s + 4096].to(device)
                q1, q2 = qnet(states[b])
                q1_sa = q1.gather(1, items[b].unsqueeze(1)).squeeze(1)
                q2_sa = q2.gather(1, items[b].unsqueeze(1)).squeeze(1)
                with torch.no_grad():
                    t1, t2 = target(next_states[b])
                    next_q = torch.minimum(t1, t2).max(dim=1).values
                    y = rewards[b] + self.gamma * (1.0 - terminals[b]) * next_q
                td = torch.nn.functional.mse_loss(q1_sa, y) + torch.nn.functional.mse_loss(q2_sa, y)
                conservative = (
                    (torch.logsumexp(q1, dim=1) - q1_sa).mean()
                    + (torch.logsumexp(q2, dim=1) - q2_sa).mean()
                )
                loss = td + self.alpha * conservative
                opt.zero_grad()
                loss.backward()
                opt.step()
                _soft_update(target, qnet, self.tau)
        self._item_emb = item_emb.weight.detach().cpu()
        self._qnet = qnet.cpu().eval()
        self._inter = inter

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        states = _SeqStates.final_states(
            dataset.interactions, self.query_column, self.item_column, self._item_emb, self._query_dim_size
        )
        with torch.no_grad():
            q1, q2 = self._qnet(states[q_ids])
            q_all = torch.minimum(q1, q2)
        scores = q_all.numpy()[:, i_ids]
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    # -- policy serialization (reference :350-392) -----------------------------
    def _save_model(self, path: str) -> None:
        import torch

        torch.save({"qnet": self._qnet.state_dict(), "item_emb": self._item_emb}, path)

    def _load_model(self, path: str) -> None:
        import torch

        blob = torch.load(path, weights_only=True)
        self._item_emb = blob["item_emb"]
        self._qnet = self._build_nets(self._item_dim_size)
        self._qnet.load_state_dict(blob["qnet"])
        self._qnet.eval()


class DDPG(Recommender):
    """Deterministic actor-critic over episode transitions with target
    networks, soft updates and Ornstein-Uhlenbeck exploration noise
    (reference experimental/models/ddpg.py semantics, native torch)."""

    def __init__(
        self,
        embedding_dim: int = 32,
        hidden_dim: int = 64,
        gamma: float = 0.9,
        tau: float = 0.05,
        epochs: int = 5,
        learning_rate: float = 1e-3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
        noise_sigma: float = 0.1,
        noise_theta: float = 0.15,
        mdp_top_k: int = 10,
    ) -> None:
        super().__init__()
        self.embedding_dim = embedding_dim
        self.hidden_dim = hidden_dim
        self.gamma = gamma
        self.tau = tau
        self.epochs = epochs
        self.learning_rate = learning_rate
        self.seed = seed
        self.device_arg = device
        self.noise_sigma = noise_sigma
        self.noise_theta = noise_theta
        self.mdp_top_k = mdp_top_k

    @property
    def _init_args(self):
        return {
            "embedding_dim": self.embedding_dim,
            "hidden_dim": self.hidden_dim,
            "gamma": self.gamma,
            "tau": self.tau,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "seed": self.seed,
            "noise_sigma": self.noise_sigma,
            "noise_theta": self.noise_theta,
            "mdp_top_k": self.mdp_top_k,
        }

    def _fit(self, dataset) -> None:
        import copy

        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions
        n_i, E, H = self._item_dim_size, self.embedding_dim, self.hidden_dim
        item_emb = torch.nn.Embedding(n_i, E)
        torch.nn.init.xavier_normal_(item_emb.weight.data)
        actor = torch.nn.Sequential(torch.nn.Linear(E, H), torch.nn.ReLU(), torch.nn.Linear(H, E))
        critic = torch.nn.Sequential(torch.nn.Linear(2 * E, H), torch.nn.ReLU(), torch.nn.Linear(H, 1))
        actor, critic = actor.to(device), critic.to(device)
        actor_t = copy.deepcopy(actor)
        critic_t = copy.deepcopy(critic)
        for m in (actor_t, critic_t):
            for p in m.parameters():
                p.requires_grad_(False)
        opt_c = torch.optim.Adam(critic.parameters(), lr=self.learning_rate)
        opt_a = torch.optim.Adam(actor.parameters(), lr=self.learning_rate)

        builder = MdpDatasetBuilder(self.mdp_top_k)
        mdp = builder.build(
            inter, self.query_column, self.item_column, self.rating_column,
            self.timestamp_column or "timestamp", seed=self.seed,
        )
        emb_cpu = item_emb.weight.detach()
        states, next_states = _SeqStates.build(mdp["users"], mdp["items"], emb_cpu)
        states, next_states = states.to(device), next_states.to(device)
        actions_emb = emb_cpu[torch.from_numpy(mdp["items"])].to(device)
        rewards = torch.from_numpy(mdp["rewards"]).to(device)
        terminals = torch.from_numpy(mdp["terminals"]).to(device)
        n = len(rewards)
        # Ornstein-Uhlenbeck noise state (exploration on the logged actions)
        ou = torch.zeros(E, device=device)
        gen = torch.Generator(device="cpu").manual_seed(self.seed or 0)
        for _ in range(self.epochs):
            perm = torch.randperm(n, generator=gen)
            for s in range(0, n, 4096):
                b = perm[s : s + 4096].to(device)
                st, nst = states[b], next_states[b]
                ou = (1.0 - self.noise_theta) * ou + self.noise_sigma * torch.randn(E, device=device)
                act = actions_emb[b] + ou
                with torch.no_grad():
                    next_q = critic_t(torch.cat([nst, actor_t(nst)], dim=-1)).squeeze(-1)
                    y = rewards[b] + self.gamma * (1.0 - terminals[b]) * next_q
                q = critic(torch.cat([st, act], dim=-1)).squeeze(-1)
                c_loss = torch.nn.functional.mse_loss(q, y)
                opt_c.zero_grad()
                c_loss.backward()
                opt_c.step()
                a_loss = -critic(torch.cat([st, actor(st)], dim=-1)).mean()
                opt_a.zero_grad()
                a_loss.backward()
                opt_a.step()
                _soft_update(actor_t, actor, self.tau)
                _soft_update(critic_t, critic, self.tau)
        self._item_emb = emb_cpu.cpu()
        self._actor = actor.cpu().eval()
        self._critic = critic.cpu().eval()

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        states = _SeqStates.final_states(
            dataset.interactions, self.query_column, self.item_column, self._item_emb, self._query_dim_size
        )
        with torch.no_grad():
            actions = self._actor(states[q_ids])  # [Q, E]
        scores = (actions @ self._item_emb[i_ids].T).numpy()
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    def _save_model(self, path: str) -> None:
        import torch

        torch.save(
            {"actor": self._actor.state_dict(), "critic": self._critic.state_dict(), "item_emb": self._item_emb},
            path,
        )

    def _load_model(self, path: str) -> None:
        import torch

        E, H = self.embedding_dim, self.hidden_dim
        blob = torch.load(path, weights_only=True)
        self._item_emb = blob["item_emb"]
        self._actor = torch.nn.Sequential(torch.nn.Linear(E, H), torch.nn.ReLU(), torch.nn.Linear(H, E))
        self._actor.load_state_dict(blob["actor"])
        self._critic = torch.nn.Sequential(torch.nn.Linear(2 * E, H), torch.nn.ReLU(), torch.nn.Linear(H, 1))
        self._critic.load_state_dict(blob["critic"])
        self._actor.eval()
