"""Neural Thompson Sampling over a Wide&Deep reward model.

Parity with reference replay/experimental/models/neural_ts.py (986 LoC):
the same capability set, MI355X/pandas-native —

* Wide&Deep model (reference :227-396): ``Wide`` linear part over one-hot
  "wide" features, ``Deep`` MLP with dropout over continuous + one-hot
  categorical features, ``EmbedModel`` user/item/cross id embeddings, and a
  dropout head combining all three.
* feature preprocessing (reference :497-614): per-side StandardScaler for
  continuous columns and one-hot encoders for categorical/wide columns,
  fitted once and reused at predict.
* training (reference :651-818): per-user batches with ``cnt_neg_samples``
  sampled negatives, WARP loss (reference :65-99) or weighted logistic
  loss (reference :47-62), AdamW + cosine LR decay.
* Thompson-sampling prediction (reference :354-362, :857-917): relevance =
  mean + exploration_coef * std over ``cnt_samples_for_predict``
  MC-dropout forward passes of the head.
* save/load (reference :920-986): encoders via joblib, weights via torch.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import numpy as np
import pandas as pd

from replay_amd.models.base_rec import HybridRecommender


def warp_loss(positive_predictions, negative_predictions, num_labels, device):
    """WARP (reference :65-99): rank-weighted hinge over sampled negatives —
    the weight is log(estimated rank) from the number of tries needed to
    find a violating negative."""
    import torch

    max_trials = negative_predictions.shape[1]
    offsets = 1 - positive_predictions.unsqueeze(1)
    scores = negative_predictions + offsets  # violation margin per trial
    tries = (scores > 0).float()
    # first violating trial index (1-based); rows without violation -> max
    first = torch.where(
        tries.any(dim=1),
        tries.argmax(dim=1) + 1,
        torch.full((scores.shape[0],), max_trials, device=device, dtype=torch.long),
    ).float()
    rank_est = torch.clamp((num_labels - 1) / first, min=1.0)
    weights = torch.log(rank_est)
    margins = torch.clamp(scores, min=0.0).max(dim=1).values
    return (weights * margins).mean()


def w_log_loss(output, target, device):
    """Class-balanced logistic loss (reference :47-62)."""
    import torch

    output = torch.sigmoid(output)
    pos = target.sum()
    n = target.shape[0]
    pos_w = (n - pos) / n
    neg_w = pos / n if pos > 0 else torch.tensor(0.5, device=device)
    weights = torch.where(target > 0.5, pos_w, neg_w)
    eps = 1e-7
    ll = target * torch.log(output + eps) + (1 - target) * torch.log(1 - output + eps)
    return -(weights * ll).mean()


class NeuralTS(HybridRecommender):
    """Neural TS bandit (reference :397): Wide&Deep reward model with
    MC-dropout posterior sampling for exploration."""

    def __init__(
        self,
        user_cols: Optional[Dict[str, List[str]]] = None,
        item_cols: Optional[Dict[str, List[str]]] = None,
        embedding_sizes: Optional[List[int]] = None,
        hidden_layers: Optional[List[int]] = None,
        wide_out_dim: int = 1,
        deep_out_dim: int = 20,
        head_dropout: float = 0.8,
        deep_dropout: float = 0.4,
        dim_head: int = 20,
        n_epochs: int = 2,
        opt_lr: float = 3e-4,
        lr_min: float = 1e-5,
        use_gpu: bool = False,
        use_warp_loss: bool = True,
        cnt_neg_samples: int = 100,
        cnt_samples_for_predict: int = 10,
        exploration_coef: float = 1.0,
        seed: Optional[int] = None,
    ) -> None:
        super().__init__()
        self.user_cols = user_cols or {"continuous_cols": [], "cat_embed_cols": [], "wide_cols": []}
        self.item_cols = item_cols or {"continuous_cols": [], "cat_embed_cols": [], "wide_cols": []}
        self.embedding_sizes = embedding_sizes or [32, 32, 64]
        self.hidden_layers = hidden_layers or [32, 20]
        self.wide_out_dim = wide_out_dim
        self.deep_out_dim = deep_out_dim
        self.head_dropout = head_dropout
        self.deep_dropout = deep_dropout
        self.dim_head = dim_head
        self.n_epochs = n_epochs
        self.opt_lr = opt_lr
        self.lr_min = lr_min
        self.use_gpu = use_gpu
        self.use_warp_loss = use_warp_loss
        self.cnt_neg_samples = cnt_neg_samples
        self.cnt_samples_for_predict = cnt_samples_for_predict
        self.exploration_coef = exploration_coef
        self.seed = seed
        self.model = None
        self._scaler_user = self._scaler_item = None
        self._ohe_user = self._ohe_item = None

    @property
    def _init_args(self):
        return {
            "user_cols": self.user_cols,
            "item_cols": self.item_cols,
            "embedding_sizes": self.embedding_sizes,
            "hidden_layers": self.hidden_layers,
            "wide_out_dim": self.wide_out_dim,
            "deep_out_dim": self.deep_out_dim,
            "head_dropout": self.head_dropout,
            "deep_dropout": self.deep_dropout,
            "dim_head": self.dim_head,
            "n_epochs": self.n_epochs,
            "opt_lr": self.opt_lr,
            "lr_min": self.lr_min,
            "use_gpu": self.use_gpu,
            "use_warp_loss": self.use_warp_loss,
            "cnt_neg_samples": self.cnt_neg_samples,
            "cnt_samples_for_predict": self.cnt_samples_for_predict,
            "exploration_coef": self.exploration_coef,
            "seed": self.seed,
        }

    # -- feature preprocessing (reference :497-614) ---------------------------
    def _fit_side(self, features: Optional[pd.DataFrame], cols: Dict[str, List[str]]):
        from sklearn.preprocessing import OneHotEncoder, StandardScaler

        scaler = ohe = None
        if features is not None:
            cont = [c for c in cols.get("continuous_cols", []) if c in features.columns]
            cats = [
                c
                for c in cols.get("cat_embed_cols", []) + cols.get("wide_cols", [])
                if c in features.columns
            ]
            if cont:
                scaler = StandardScaler().fit(features[cont].to_numpy(dtype=np.float64))
            if cats:
                ohe = OneHotEncoder(handle_unknown="ignore", sparse_output=False).fit(
                    features[cats]
                )
        return scaler, ohe

    def _transform_side(self, features, cols, scaler, ohe, ids, id_col):
        """Returns (continuous [n, c], onehot [n, o]) aligned to ``ids``."""
        n = len(ids)
        cont = np.zeros((n, 0), dtype=np.float32)
        cat = np.zeros((n, 0), dtype=np.float32)
        if features is None:
            return cont, cat
        feats = features.set_index(id_col).reindex(ids)
        cont_cols = [c for c in cols.get("continuous_cols", []) if c in features.columns]
        cat_cols = [
            c
            for c in cols.get("cat_embed_cols", []) + cols.get("wide_cols", [])
            if c in features.columns
        ]
        if cont_cols and scaler is not None:
            cont = scaler.transform(
                feats[cont_cols].fillna(0.0).to_numpy(dtype=np.float64)
            ).astype(np.float32)
        if cat_cols and ohe is not None:
            cat = ohe.transform(feats[cat_cols].fillna("<na>")).astype(np.float32)
        return cont, cat

    def _pair_features(self, users, items, user_features, item_features):
        uc, uo = self._transform_side(
            user_features, self.user_cols, self._scaler_user, self._ohe_user, users, self.query_column
        )
        ic, io = self._transform_side(
            item_features, self.item_cols, self._scaler_item, self._ohe_item, items, self.item_column
        )
        cont = np.concatenate([uc, ic], axis=1)
        wide = np.concatenate([uo, io], axis=1)
        return cont, wide

    # -- model (reference :227-396) --------------------------------------------
    def _build_model(self, n_wide, n_cont, n_users, n_items):
        import torch

        ue, ie, ce = self.embedding_sizes
        hid = self.hidden_layers
        deep_in = n_cont + n_wide

        class WideDeep(torch.nn.Module):
            def __init__(inner):
                super().__init__()
                inner.wide = torch.nn.Linear(max(n_wide, 1), self.wide_out_dim)
                layers = []
                dims = [max(deep_in, 1)] + hid
                for a, b in zip(dims[:-1], dims[1:]):
                    layers += [torch.nn.Linear(a, b), torch.nn.ReLU(), torch.nn.Dropout(self.deep_dropout)]
                layers += [torch.nn.Linear(dims[-1], self.deep_out_dim), torch.nn.Dropout(self.deep_dropout)]
                inner.deep = torch.nn.Sequential(*layers)
                inner.u_emb = torch.nn.Embedding(n_users, ue)
                inner.i_emb = torch.nn.Embedding(n_items, ie)
                inner.u_cross = torch.nn.Embedding(n_users, ce)
                inner.i_cross = torch.nn.Embedding(n_items, ce)
                inner.head = torch.nn.Linear(
                    self.wide_out_dim + self.deep_out_dim + ue + ie + 1, self.dim_head
                )
                inner.out = torch.nn.Linear(self.dim_head, 1)
                inner.head_dropout_p = self.head_dropout

            def features(inner, wide, cont, users, items):
                import torch as t

                wide_in = wide if wide.shape[1] else t.zeros(wide.shape[0], 1, device=wide.device)
                deep_in_t = t.cat([cont, wide], dim=1)
                if not deep_in_t.shape[1]:
                    deep_in_t = t.zeros(wide.shape[0], 1, device=wide.device)
                cross = (inner.u_cross(users) * inner.i_cross(items)).sum(-1, keepdim=True)
                parts = [
                    inner.wide(wide_in),
                    inner.deep(deep_in_t),
                    inner.u_emb(users),
                    inner.i_emb(items),
                    cross,
                ]
                return inner.head(t.cat(parts, dim=1))

            def head_sample(inner, feats):
                """One MC-dropout head sample (reference forward_dropout)."""
                import torch as t

                return inner.out(
                    t.nn.functional.dropout(feats, p=inner.head_dropout_p, training=True)
                ).squeeze(-1)

            def forward(inner, wide, cont, users, items):
                return inner.head_sample(inner.features(wide, cont, users, items))

        return WideDeep()

    # -- fit (reference :651-818) ----------------------------------------------
    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.seed or 0)
        rng = np.random.default_rng(self.seed)
        device = torch.device("cuda" if self.use_gpu and torch.cuda.is_available() else "cpu")
        self._device = device
        inter = dataset.interactions
        user_features = dataset.query_features
        item_features = dataset.item_features
        self._scaler_user, self._ohe_user = self._fit_side(user_features, self.user_cols)
        self._scaler_item, self._ohe_item = self._fit_side(item_features, self.item_cols)
        self._user_features = user_features
        self._item_features = item_features

        users = inter[self.query_column].to_numpy(dtype=np.int64)
        items = inter[self.item_column].to_numpy(dtype=np.int64)
        n_users, n_items = self._query_dim_size, self._item_dim_size
        all_items = np.arange(n_items)

        # per-user positive sets + sampled negatives (reference :120-225)
        pos_by_user = pd.Series(items).groupby(pd.Series(users)).apply(np.asarray).to_dict()
        rows_u, rows_i, rows_y = [], [], []
        for u, pos in pos_by_user.items():
            rows_u.append(np.full(len(pos), u))
            rows_i.append(pos)
            rows_y.append(np.ones(len(pos)))
            if self.cnt_neg_samples > 0:
                negs = rng.choice(all_items, size=min(self.cnt_neg_samples, n_items), replace=False)
                negs = negs[~np.isin(negs, pos)]
                rows_u.append(np.full(len(negs), u))
                rows_i.append(negs)
                rows_y.append(np.zeros(len(negs)))
        u_all = np.concatenate(rows_u)
        i_all = np.concatenate(rows_i)
        y_all = np.concatenate(rows_y)

        cont, wide = self._pair_features(u_all, i_all, user_features, item_features)
        self._n_wide, self._n_cont = wide.shape[1], cont.shape[1]
        self.model = self._build_model(self._n_wide, self._n_cont, n_users, n_items).to(device)
        opt = torch.optim.AdamW(self.model.parameters(), lr=self.opt_lr)
        sched = torch.optim.lr_scheduler.CosineAnnealingLR(
            opt, T_max=max(self.n_epochs, 1), eta_min=self.lr_min
        )
        t = lambda x, dt=torch.float32: torch.as_tensor(x, dtype=dt, device=device)  # noqa: E731
        wide_t, cont_t = t(wide), t(cont)
        u_t, i_t = t(u_all, torch.long), t(i_all, torch.long)
        y_t = t(y_all)
        order = np.arange(len(u_all))
        self.model.train()
        for _ in range(self.n_epochs):
            rng.shuffle(order)
            for s in range(0, len(order), 8192):
                b = torch.as_tensor(order[s : s + 8192], device=device)
                preds = self.model(wide_t[b], cont_t[b], u_t[b], i_t[b])
                yb = y_t[b]
                if self.use_warp_loss:
                    pos_mask = yb > 0.5
                    if bool(pos_mask.any()) and bool((~pos_mask).any()):
                        pos_p = preds[pos_mask]
                        neg_p = preds[~pos_mask]
                        k = min(len(neg_p), 32)
                        idx = torch.randint(0, len(neg_p), (len(pos_p), k), device=device)
                        loss = warp_loss(pos_p, neg_p[idx], n_items, device)
                    else:
                        loss = w_log_loss(preds, yb, device)
                else:
                    loss = w_log_loss(preds, yb, device)
                opt.zero_grad()
                loss.backward()
                opt.step()
            sched.step()
        self.model.eval()

    # -- Thompson-sampling predict (reference :857-917) -------------------------
    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        user_features = dataset.query_features if dataset is not None else self._user_features
        item_features = dataset.item_features if dataset is not None else self._item_features
        scores = np.zeros((len(q_ids), len(i_ids)), dtype=np.float32)
        with torch.no_grad():
            for qi, q in enumerate(q_ids):
                u_rep = np.full(len(i_ids), q)
                cont, wide = self._pair_features(u_rep, i_ids, user_features, item_features)
                wt = torch.as_tensor(wide, device=self._device)
                ct = torch.as_tensor(cont, device=self._device)
                ut = torch.as_tensor(u_rep, dtype=torch.long, device=self._device)
                it = torch.as_tensor(i_ids, dtype=torch.long, device=self._device)
                feats = self.model.features(wt, ct, ut, it)
                samples = torch.stack(
                    [self.model.head_sample(feats) for _ in range(self.cnt_samples_for_predict)]
                )
                rel = samples.mean(0) + self.exploration_coef * samples.var(0).clamp(min=0).sqrt()
                scores[qi] = rel.cpu().numpy()
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    # -- save/load (reference :920-986) -----------------------------------------
    def model_save(self, dir_name: str) -> None:
        import joblib
        import torch

        os.makedirs(dir_name, exist_ok=True)
        joblib.dump(
            {
                "scaler_user": self._scaler_user,
                "ohe_user": self._ohe_user,
                "scaler_item": self._scaler_item,
                "ohe_item": self._ohe_item,
                "n_wide": self._n_wide,
                "n_cont": self._n_cont,
                "n_users": self._query_dim_size,
                "n_items": self._item_dim_size,
            },
            os.path.join(dir_name, "encoders.joblib"),
        )
        torch.save(self.model.state_dict(), os.path.join(dir_name, "model_weights.pth"))

    def model_load(self, dir_name: str) -> None:
        import joblib
        import torch

        blob = joblib.load(os.path.join(dir_name, "encoders.joblib"))
        self._scaler_user = blob["scaler_user"]
        self._ohe_user = blob["ohe_user"]
        self._scaler_item = blob["scaler_item"]
        self._ohe_item = blob["ohe_item"]
        self._n_wide, self._n_cont = blob["n_wide"], blob["n_cont"]
        self.model = self._build_model(
            self._n_wide, self._n_cont, blob["n_users"], blob["n_items"]
        )
        self.model.load_state_dict(
            torch.load(os.path.join(dir_name, "model_weights.pth"), weights_only=True)
        )
        self.model.eval()
        self._device = "cpu"
