"""Binary cross-entropy losses (reference replay/nn/loss/bce.py: BCE:10 full,
BCESampled:98 sampled positives+negatives)."""

from __future__ import annotations

from typing import Optional

import torch

from .base import LossBase, SampledLossBase


class BCE(LossBase):
    """Full BCE: 1 at the positive label, 0 elsewhere over the catalog."""

    def forward(
        self,
        embeddings: torch.Tensor,
        positive_labels: torch.Tensor,
        padding_mask: torch.Tensor,
        target_padding_mask: Optional[torch.Tensor] = None,
        negative_labels: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        logits = self.logits_callback(embeddings).float()  # [B, L, V]
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        target = torch.zeros_like(logits)
        safe = positive_labels.clamp(min=0)
        target.scatter_(-1, safe.unsqueeze(-1), 1.0)
        per_elem = torch.nn.functional.binary_cross_entropy_with_logits(logits, target, reduction="none")
        valid = mask.unsqueeze(-1).to(per_elem.dtype)
        # reference bce.py: SUM over the catalog axis, MEAN over valid
        # positions (not over positions x catalog)
        return (per_elem * valid).sum() / valid.sum().clamp(min=1e-12)


class BCESampled(SampledLossBase):
    """-logsigmoid(pos) - sum logsigmoid(-neg) over sampled negatives."""

    def forward(
        self,
        embeddings: torch.Tensor,
        positive_labels: torch.Tensor,
        padding_mask: torch.Tensor,
        target_padding_mask: Optional[torch.Tensor] = None,
        negative_labels: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if negative_labels is None:
            raise ValueError("BCESampled requires negative_labels")
        pos, neg = self.get_sampled_logits(embeddings, positive_labels, negative_labels)
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        pos_term = torch.nn.functional.logsigmoid(pos.float()).squeeze(-1)
        neg_valid = torch.isfinite(neg)
        neg_f = neg.float().masked_fill(~neg_valid, 0.0)
        neg_term = (torch.nn.functional.logsigmoid(-neg_f) * neg_valid).sum(-1) / neg_valid.sum(-1).clamp(min=1)
        per_pos = -(pos_term + neg_term)
        valid = mask.to(per_pos.dtype)
        if weights is not None:
            valid = valid * weights
        return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)
