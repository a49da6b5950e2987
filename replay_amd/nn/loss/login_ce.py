"""InfoNCE losses.

Parity with reference replay/nn/loss/login_ce.py (LogInCE/LogInCEBase:19-92 —
positives appear IN the logit set, multi-positive support) and logout_ce.py
(LogOutCE:10 — positives excluded from the denominator set).
"""

from __future__ import annotations

from typing import Optional

import torch

from .base import SampledLossBase


class LogInCE(SampledLossBase):
    """-log( exp(pos) / (exp(pos) + sum exp(neg)) ), positives in-logits."""

    def __init__(self, temperature: float = 1.0) -> None:
        super().__init__()
        self.temperature = temperature

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
            # in-batch negatives: the batch's own positives act as the
            # negative pool (reference login_ce.py LogInCE vs LogInCESampled:
            # the only difference is the negative set)
            negative_labels = positive_labels[padding_mask].reshape(-1)
        if self.temperature == 1.0:
            from replay_amd.ops.sampled_ce import can_fuse_sampled_ce, fused_sampled_ce_parts

            if can_fuse_sampled_ce(embeddings, negative_labels, self.logits_callback):
                # K9 fused path (see ops/sampled_ce.py): positives-in-logits
                # InfoNCE is softplus(lse_neg - pos) once collisions are
                # excluded, so the shared-pool LSE kernel covers it
                pos_logit, lse_neg, _ = fused_sampled_ce_parts(
                    embeddings, positive_labels, negative_labels, self.logits_callback
                )
                per_pos = torch.nn.functional.softplus(lse_neg - pos_logit)
                mask = target_padding_mask if target_padding_mask is not None else padding_mask
                valid = mask.to(per_pos.dtype)
                if weights is not None:
                    valid = valid * weights
                return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)
        pos, neg = self.get_sampled_logits(embeddings, positive_labels, negative_labels)
        logits = torch.cat([pos, neg], dim=-1).float() / self.temperature
        lse = torch.logsumexp(logits, dim=-1)
        per_pos = lse - pos.squeeze(-1).float() / self.temperature
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        valid = mask.to(per_pos.dtype)
        if weights is not None:
            valid = valid * weights
        return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)


class LogInCESampled(LogInCE):
    """LogInCE over explicitly sampled negatives (reference login_ce.py:241;
    ``LogInCE`` itself defaults to the in-batch pool when none are given)."""

    def forward(self, *args, **kwargs) -> torch.Tensor:
        negs = kwargs.get("negative_labels")
        if negs is None and len(args) < 5:
            raise ValueError("LogInCESampled requires negative_labels")
        return super().forward(*args, **kwargs)


class LogOutCE(SampledLossBase):
    """InfoNCE with positives OUT of the denominator (reference logout_ce.py:10)."""

    def __init__(self, temperature: float = 1.0) -> None:
        super().__init__()
        self.temperature = temperature

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
            raise ValueError("LogOutCE requires negative_labels")
        pos, neg = self.get_sampled_logits(embeddings, positive_labels, negative_labels)
        pos_f = pos.squeeze(-1).float() / self.temperature
        neg_f = neg.float() / self.temperature
        lse_neg = torch.logsumexp(neg_f, dim=-1)
        # softplus(lse_neg - pos) = -log(exp(pos) / (exp(pos) + sum exp(neg)))
        per_pos = torch.nn.functional.softplus(lse_neg - pos_f)
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        valid = mask.to(per_pos.dtype)
        if weights is not None:
            valid = valid * weights
        return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)
