"""MultVAE (reference replay/experimental/models/mult_vae.py, 333 LoC):
variational autoencoder with multinomial likelihood over each user's item
vector (Liang et al. 2018).  Torch-native; trains on GPU when available."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
from scipy.sparse import csr_matrix

from replay_amd.models.base_rec import Recommender


class MultVAE(Recommender):
    def __init__(
        self,
        learning_rate: float = 0.01,
        epochs: int = 100,
        latent_dim: int = 200,
        hidden_dim: int = 600,
        dropout: float = 0.3,
        anneal: float = 0.1,
        l2_reg: float = 0.0,
        factor: float = 0.8,
        patience: int = 3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
    ) -> None:
        super().__init__()
        self.learning_rate = learning_rate
        self.epochs = epochs
        self.latent_dim = latent_dim
        self.hidden_dim = hidden_dim
        self.dropout = dropout
        self.anneal = anneal
        self.l2_reg = l2_reg
        self.seed = seed
        self.device_arg = device
        self._net = None

    @property
    def _init_args(self):
        return {
            "learning_rate": self.learning_rate,
            "epochs": self.epochs,
            "latent_dim": self.latent_dim,
            "hidden_dim": self.hidden_dim,
            "dropout": self.dropout,
            "anneal": self.anneal,
            "seed": self.seed,
        }

    _search_space = {
        "latent_dim": {"type": "int", "args": [64, 400]},
        "anneal": {"type": "uniform", "args": [0.0, 1.0]},
    }

    def _build_net(self, n_items, torch):
        class Net(torch.nn.Module):
            def __init__(self, n_items, hidden, latent, dropout):
                super().__init__()
                self.encoder = torch.nn.Sequential(
                    torch.nn.Dropout(dropout),
                    torch.nn.Linear(n_items, hidden),
                    torch.nn.Tanh(),
                )
                self.mu = torch.nn.Linear(hidden, latent)
                self.logvar = torch.nn.Linear(hidden, latent)
                self.decoder = torch.nn.Sequential(
                    torch.nn.Linear(latent, hidden),
                    torch.nn.Tanh(),
                    torch.nn.Linear(hidden, n_items),
                )

            def forward(self, x):
                h = self.encoder(torch.nn.functional.normalize(x, dim=-1))
                mu, logvar = self.mu(h), self.logvar(h)
                if self.training:
                    z = mu + torch.randn_like(mu) * torch.exp(0.5 * logvar)
                else:
                    z = mu
                return self.decoder(z), mu, logvar

        return Net(n_items, self.hidden_dim, self.latent_dim, self.dropout)

    def _user_matrix(self, dataset) -> csr_matrix:
        inter = dataset.interactions
        rows = inter[self.query_column].to_numpy(dtype=np.int64)
        cols = inter[self.item_column].to_numpy(dtype=np.int64)
        return csr_matrix(
            (np.ones(len(inter)), (rows, cols)), shape=(self._query_dim_size, self._item_dim_size)
        )

    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        mat = self._user_matrix(dataset)
        self._net = self._build_net(mat.shape[1], torch).to(device)
        opt = torch.optim.Adam(self._net.parameters(), lr=self.learning_rate, weight_decay=self.l2_reg)
        dense = torch.from_numpy(mat.toarray().astype(np.float32))
        batch = 256
        self._net.train()
        for _ in range(self.epochs):
            perm = torch.randperm(dense.shape[0])
            for s in range(0, len(perm), batch):
                x = dense[perm[s : s + batch]].to(device)
                logits, mu, logvar = self._net(x)
                log_softmax = torch.nn.functional.log_softmax(logits, dim=-1)
                neg_ll = -(log_softmax * x).sum(-1).mean()
                kld = -0.5 * (1 + logvar - mu.pow(2) - logvar.exp()).sum(-1).mean()
                loss = neg_ll + self.anneal * kld
                opt.zero_grad()
                loss.backward()
                opt.step()
        self._net.eval()
        self._device = device
        self._train_matrix = dense

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        with torch.no_grad():
            x = self._train_matrix[q_ids].to(self._device)
            logits, _, _ = self._net(x)
        scores = logits.cpu().numpy()[:, i_ids]
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    def _save_model(self, path: str) -> None:
        import torch

        torch.save({"net": self._net.state_dict(), "train": self._train_matrix}, f"{path}/vae.pt")

    def _load_model(self, path: str) -> None:
        import torch

        state = torch.load(f"{path}/vae.pt", weights_only=False)
        self._net = self._build_net(self._item_dim_size, torch)
        self._net.load_state_dict(state["net"])
        self._net.eval()
        self._train_matrix = state["train"]
        self._device = "cpu"
