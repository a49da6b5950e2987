"""Word2Vec item-embedding recommender.

Parity with reference Word2VecRec (replay/models/word2vec.py:22), which wraps
``pyspark.ml.feature.Word2Vec`` over per-user item sequences; the query vector
is the (optionally idf-weighted) mean of its items' vectors.  Here the
skip-gram-negative-sampling model is trained in torch (runs on the MI355X when
available; CPU otherwise) — no JVM, no gensim.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import pandas as pd

from .base_rec import ItemVectorModel, Recommender


class Word2VecRec(Recommender, ItemVectorModel):
    def __init__(
        self,
        rank: int = 100,
        min_count: int = 5,
        step_size: float = 0.025,
        max_iter: int = 1,
        window_size: int = 1,
        use_idf: bool = False,
        seed: Optional[int] = None,
        num_negatives: int = 5,
        batch_size: int = 8192,
        device: Optional[str] = None,
    ) -> None:
        super().__init__()
        self.rank = rank
        self.min_count = min_count
        self.step_size = step_size
        self.max_iter = max_iter
        self.window_size = window_size
        self.use_idf = use_idf
        self.seed = seed
        self.num_negatives = num_negatives
        self.batch_size = batch_size
        self.device = device
        self.item_vectors: Optional[np.ndarray] = None
        self.idf: Optional[np.ndarray] = None

    @property
    def _init_args(self) -> Dict:
        return {
            "rank": self.rank,
            "min_count": self.min_count,
            "step_size": self.step_size,
            "max_iter": self.max_iter,
            "window_size": self.window_size,
            "use_idf": self.use_idf,
            "seed": self.seed,
        }

    _search_space = {
        "rank": {"type": "int", "args": [50, 300]},
        "window_size": {"type": "int", "args": [1, 100]},
        "use_idf": {"type": "categorical", "args": [True, False]},
    }

    def _save_model(self, path: str) -> None:
        np.savez(f"{path}/w2v.npz", vectors=self.item_vectors, idf=self.idf)

    def _load_model(self, path: str) -> None:
        data = np.load(f"{path}/w2v.npz")
        self.item_vectors, self.idf = data["vectors"], data["idf"]

    def _fit(self, dataset) -> None:
        import torch

        inter = dataset.interactions.sort_values(
            [self.query_column, self.timestamp_column]
            if self.timestamp_column in dataset.interactions.columns
            else [self.query_column],
            kind="stable",
        )
        sequences = inter.groupby(self.query_column)[self.item_column].apply(np.asarray)
        n_items = self._item_dim_size

        # idf over user-documents
        doc_freq = inter.groupby(self.item_column)[self.query_column].nunique()
        idf = np.zeros(n_items)
        idf[doc_freq.index.to_numpy()] = np.log((len(sequences) + 1) / (doc_freq.to_numpy() + 1)) + 1
        self.idf = idf

        # build skip-gram pairs within window
        centers, contexts = [], []
        for seq in sequences:
            L = len(seq)
            for w in range(1, self.window_size + 1):
                if L > w:
                    centers.append(seq[:-w])
                    contexts.append(seq[w:])
                    centers.append(seq[w:])
                    contexts.append(seq[:-w])
        if not centers:
            self.item_vectors = np.zeros((n_items, self.rank), dtype=np.float32)
            return
        centers = np.concatenate(centers)
        contexts = np.concatenate(contexts)

        device = self.device or ("cuda" if torch.cuda.is_available() else "cpu")
        gen = torch.Generator(device="cpu").manual_seed(self.seed or 0)
        emb_in = torch.nn.Embedding(n_items, self.rank, device=device)
        emb_out = torch.nn.Embedding(n_items, self.rank, device=device)
        torch.nn.init.normal_(emb_in.weight, std=0.5 / self.rank, generator=None)
        torch.nn.init.zeros_(emb_out.weight)
        opt = torch.optim.Adam(list(emb_in.parameters()) + list(emb_out.parameters()), lr=self.step_size)

        c_t = torch.from_numpy(centers.astype(np.int64))
        x_t = torch.from_numpy(contexts.astype(np.int64))
        n = len(c_t)
        for _ in range(self.max_iter):
            perm = torch.randperm(n, generator=gen)
            for s in range(0, n, self.batch_size):
                b = perm[s : s + self.batch_size]
                cb = c_t[b].to(device)
                xb = x_t[b].to(device)
                neg = torch.randint(0, n_items, (len(b), self.num_negatives), generator=gen).to(device)
                vc = emb_in(cb)  # [B, f]
                vx = emb_out(xb)  # [B, f]
                vn = emb_out(neg)  # [B, neg, f]
                pos_logit = (vc * vx).sum(-1)
                neg_logit = torch.einsum("bf,bnf->bn", vc, vn)
                loss = (
                    torch.nn.functional.softplus(-pos_logit).mean()
                    + torch.nn.functional.softplus(neg_logit).mean()
                )
                opt.zero_grad(set_to_none=True)
                loss.backward()
                opt.step()
        self.item_vectors = emb_in.weight.detach().cpu().numpy()

    def _query_vectors(self, dataset, q_ids: np.ndarray) -> np.ndarray:
        inter = dataset.interactions
        weights = self.idf if self.use_idf else np.ones(len(self.idf))
        vecs = np.zeros((len(q_ids), self.rank), dtype=np.float32)
        grouped = inter[inter[self.query_column].isin(set(q_ids.tolist()))].groupby(self.query_column)[
            self.item_column
        ]
        seqs = grouped.apply(np.asarray).to_dict()
        for i, q in enumerate(q_ids):
            seq = seqs.get(q)
            if seq is None or not len(seq):
                continue
            w = weights[seq]
            vecs[i] = (self.item_vectors[seq] * w[:, None]).sum(0) / max(len(seq), 1)
        return vecs

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        qv = self._query_vectors(dataset, q_ids)
        scores = qv @ self.item_vectors[i_ids].T
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))

    def _get_item_vectors(self) -> pd.DataFrame:
        return pd.DataFrame(
            {
                self.item_column: np.arange(len(self.item_vectors)),
                "item_vector": list(self.item_vectors),
            }
        )
