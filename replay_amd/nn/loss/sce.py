"""Scalable (bucketed) cross-entropy.

Parity with reference ScalableCrossEntropyLoss
(replay/models/nn/loss/sce.py:27-124): random-projection buckets (n_b, hd);
per bucket the top-``bucket_size_x`` hidden states and top-``bucket_size_y``
classes by projection score; CE computed on the (n_b, bs_x, bs_y+1) slice with
the true positive appended; scatter-amax accumulates each position's best
bucket loss.

MI355X note: the bucket GEMMs are MFMA-shaped (K11 in SURVEY §2.12) and reuse
the catalog-scoring tiles of K7.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from .base import LossBase


class ScalableCrossEntropyLoss(LossBase):
    def __init__(
        self,
        n_buckets: int = 32,
        bucket_size_x: Optional[int] = None,
        bucket_size_y: Optional[int] = None,
        mix_x: bool = False,
    ) -> None:
        super().__init__()
        self.n_buckets = n_buckets
        self.bucket_size_x = bucket_size_x
        self.bucket_size_y = bucket_size_y
        self.mix_x = mix_x

    def forward(
        self,
        embeddings: torch.Tensor,  # [B, L, E]
        positive_labels: torch.Tensor,  # [B, L]
        padding_mask: torch.Tensor,
        target_padding_mask: Optional[torch.Tensor] = None,
        negative_labels: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
        item_embeddings: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if item_embeddings is None:
            item_embeddings = self.logits_callback.get_item_weights()  # [V, E]
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        B, L, E = embeddings.shape
        x = embeddings.reshape(-1, E)  # [N, E]
        y = positive_labels.reshape(-1).clamp(min=0)  # [N]
        valid = mask.reshape(-1)
        x = x[valid]
        y = y[valid]
        N = x.shape[0]
        if N == 0:
            return embeddings.sum() * 0.0
        V = item_embeddings.shape[0]
        n_b = self.n_buckets
        bs_x = self.bucket_size_x or max(1, int(2 * N / n_b))
        bs_y = self.bucket_size_y or max(1, int(2 * V / n_b))
        bs_x = min(bs_x, N)
        bs_y = min(bs_y, V)

        w = torch.randn(n_b, E, device=x.device, dtype=torch.float32) / math.sqrt(E)
        xf = x.float()
        ef = item_embeddings.float()
        px = w @ xf.T  # [n_b, N]
        py = w @ ef.T  # [n_b, V]
        top_x = px.topk(bs_x, dim=-1).indices  # [n_b, bs_x]
        top_y = py.topk(bs_y, dim=-1).indices  # [n_b, bs_y]

        xb = xf[top_x]  # [n_b, bs_x, E]
        yb = ef[top_y]  # [n_b, bs_y, E]
        logits = torch.einsum("bxe,bye->bxy", xb, yb)  # [n_b, bs_x, bs_y]
        labels_b = y[top_x]  # [n_b, bs_x]
        pos_emb = ef[labels_b]  # [n_b, bs_x, E]
        pos_logit = (xb * pos_emb).sum(-1, keepdim=True)  # [n_b, bs_x, 1]
        # mask in-bucket copies of the positive class
        same = top_y.unsqueeze(1) == labels_b.unsqueeze(-1)  # [n_b, bs_x, bs_y]
        logits = logits.masked_fill(same, float("-inf"))
        full = torch.cat([pos_logit, logits], dim=-1)  # [n_b, bs_x, bs_y+1]
        per = torch.nn.functional.cross_entropy(
            full.reshape(-1, full.shape[-1]),
            torch.zeros(full.shape[0] * full.shape[1], dtype=torch.long, device=x.device),
            reduction="none",
        ).reshape(n_b, bs_x)

        # scatter-amax: each position keeps its hardest (max-loss) bucket view
        acc = torch.full((N,), float("-inf"), device=x.device)
        acc = acc.scatter_reduce(0, top_x.reshape(-1), per.reshape(-1), reduce="amax", include_self=True)
        covered = torch.isfinite(acc)
        if not covered.any():
            return embeddings.sum() * 0.0
        return acc[covered].mean()
