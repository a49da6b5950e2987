"""DT4Rec: decision-transformer recommender.

Parity with reference replay/experimental/models/dt4rec/ (gpt1.py 401 LoC +
trainer.py + utils.py — a GPT-1 decision transformer over (return-to-go,
state, action) triples).  Re-composed from this framework's causal
transformer blocks: per step THREE interleaved tokens
[rtg_t, state_t, action_t] (state = mean embedding of the last
``memory_size`` items, reference GPTConfig.memory_size), sliding-window
trajectory samples over long histories (reference
StateActionReturnDataset), warmup LR schedule (reference WarmUpScheduler),
and high-return conditioning at inference.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from replay_amd.models.base_rec import Recommender


class DT4Rec(Recommender):
    def __init__(
        self,
        embedding_dim: int = 64,
        num_blocks: int = 2,
        num_heads: int = 2,
        max_sequence_length: int = 30,
        epochs: int = 3,
        learning_rate: float = 1e-3,
        seed: Optional[int] = None,
        device: Optional[str] = None,
        batch_size: int = 128,
        memory_size: int = 3,
        warmup_steps: int = 100,
    ) -> None:
        super().__init__()
        self.memory_size = memory_size
        self.warmup_steps = warmup_steps
        self.embedding_dim = embedding_dim
        self.num_blocks = num_blocks
        self.num_heads = num_heads
        self.max_sequence_length = max_sequence_length
        self.epochs = epochs
        self.learning_rate = learning_rate
        self.seed = seed
        self.device_arg = device
        self.batch_size = batch_size
        self._net = None

    @property
    def _init_args(self):
        return {
            "embedding_dim": self.embedding_dim,
            "num_blocks": self.num_blocks,
            "num_heads": self.num_heads,
            "max_sequence_length": self.max_sequence_length,
            "epochs": self.epochs,
            "learning_rate": self.learning_rate,
            "seed": self.seed,
            "memory_size": self.memory_size,
            "warmup_steps": self.warmup_steps,
        }

    def _build_net(self, n_items, torch):
        from replay_amd.nn.mask import MaskSpec
        from replay_amd.nn.sequential.sasrec.transformer import SasRecTransformerLayer

        E, H, B_, L = self.embedding_dim, self.num_heads, self.num_blocks, self.max_sequence_length

        mem = self.memory_size

        class GPT(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.item_emb = torch.nn.Embedding(n_items + 1, E, padding_idx=n_items)
                self.rtg_proj = torch.nn.Linear(1, E)
                self.state_proj = torch.nn.Linear(E, E)  # over the memory mean
                self.pos = torch.nn.Embedding(3 * L, E)
                self.encoder = SasRecTransformerLayer(E, H, B_, dropout=0.1, activation="gelu")
                self.head = torch.nn.Linear(E, n_items)

            def forward(self, items, rtg, mask):
                # per step THREE tokens: [rtg_t, state_t, action_t] where
                # state_t = mean of the previous `mem` item embeddings
                # (reference gpt1.py memory_size observation)
                bsz, seq = items.shape
                e_items = self.item_emb(items)
                kernel = torch.ones(mem, device=items.device) / mem
                padded = torch.cat(
                    [torch.zeros(bsz, mem, e_items.shape[-1], device=items.device), e_items[:, :-1]],
                    dim=1,
                )
                state = torch.stack(
                    [padded[:, t : t + mem].mean(dim=1) for t in range(seq)], dim=1
                )
                del kernel
                e_state = self.state_proj(state)
                e_rtg = self.rtg_proj(rtg.unsqueeze(-1))
                tokens = torch.stack([e_rtg, e_state, e_items], dim=2).reshape(bsz, 3 * seq, E)
                tok_mask = mask.repeat_interleave(3, dim=1)
                positions = torch.arange(3 * seq, device=items.device)
                tokens = tokens + self.pos(positions)[None]
                spec = MaskSpec(tok_mask, H, True, self.training)
                hidden = self.encoder(tokens, attn_mask=spec, padding_mask=tok_mask)
                # predict item_t from the state token of step t
                return self.head(hidden[:, 1::3])

        return GPT()

    def _fit(self, dataset) -> None:
        import torch

        torch.manual_seed(self.seed or 0)
        device = self.device_arg or ("cuda" if torch.cuda.is_available() else "cpu")
        inter = dataset.interactions.sort_values(
            [self.query_column, self.timestamp_column]
            if self.timestamp_column in dataset.interactions.columns
            else [self.query_column]
        )
        n_items = self._item_dim_size
        L = self.max_sequence_length
        seqs, rtgs = [], []
        last_rows = {}
        for q, g in inter.groupby(self.query_column):
            full_items = g[self.item_column].to_numpy(dtype=np.int64)
            full_rewards = (
                g[self.rating_column].to_numpy(dtype=np.float32)
                if self.rating_column in g.columns
                else np.ones(len(full_items), dtype=np.float32)
            )
            full_rtg = full_rewards[::-1].cumsum()[::-1].copy()  # return-to-go
            # sliding trajectory windows over long histories (reference
            # StateActionReturnDataset: every length-L window is a sample)
            starts = list(range(0, max(1, len(full_items) - L + 1), max(1, L // 2)))
            if starts[-1] != max(0, len(full_items) - L):
                starts.append(max(0, len(full_items) - L))
            for st in dict.fromkeys(starts):  # dedupe, keep order
                items = full_items[st : st + L]
                rtg = full_rtg[st : st + L]
                pad = L - len(items)
                seqs.append(np.concatenate([np.full(pad, n_items), items]))
                rtgs.append(np.concatenate([np.zeros(pad, dtype=np.float32), rtg]))
            last_rows[int(q)] = len(seqs) - 1  # the user's final window
        items_t = torch.from_numpy(np.stack(seqs))
        rtg_t = torch.from_numpy(np.stack(rtgs))
        mask_t = items_t != n_items
        self._net = self._build_net(n_items, torch).to(device)
        opt = torch.optim.AdamW(self._net.parameters(), lr=self.learning_rate, betas=(0.9, 0.95))
        warm = max(1, self.warmup_steps)
        sched = torch.optim.lr_scheduler.LambdaLR(opt, lambda t: min(1.0, (t + 1) / warm))
        self._net.train()
        for _ in range(self.epochs):
            perm = torch.randperm(len(items_t))
            for s in range(0, len(perm), self.batch_size):
                b = perm[s : s + self.batch_size]
                it, rt, mk = items_t[b].to(device), rtg_t[b].to(device), mask_t[b].to(device)
                logits = self._net(it, rt, mk)  # predict item_t from rtg_t
                labels = it.masked_fill(~mk, -100)
                loss = torch.nn.functional.cross_entropy(
                    logits.reshape(-1, logits.shape[-1]), labels.reshape(-1), ignore_index=-100
                )
                opt.zero_grad()
                loss.backward()
                opt.step()
                sched.step()
        self._net.eval()
        self._device = device
        self._train_items = items_t
        self._train_rtg = rtg_t
        self._train_mask = mask_t
        self._query_row = last_rows  # each user's final window (inference context)

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        import torch

        q_ids = queries[self.query_column].to_numpy(dtype=np.int64)
        i_ids = items[self.item_column].to_numpy(dtype=np.int64)
        rows = [self._query_row.get(int(q), 0) for q in q_ids]
        it = self._train_items[rows].to(self._device)
        # condition on a high desired return (standard DT inference trick)
        rt = torch.full_like(self._train_rtg[rows], float(self._train_rtg.max())).to(self._device)
        mk = self._train_mask[rows].to(self._device)
        with torch.no_grad():
            logits = self._net(it, rt, mk)
        from replay_amd.nn.utils import last_valid_index

        last = last_valid_index(mk)
        scores = logits[torch.arange(len(rows), device=self._device), last].cpu().numpy()[:, i_ids]
        return self._recs_from_scores(scores, q_ids, i_ids, min(k, len(i_ids)))
