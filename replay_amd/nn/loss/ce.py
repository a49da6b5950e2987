"""Cross-entropy losses.

Parity with reference replay/nn/loss/ce.py (CE:10 full-softmax over [B,L,V]
with pad ignore :49-82; CEWeighted:84; CESampled:146 sampled
positives+negatives with collision masking; CESampledWeighted:252).

MI355X note: full-softmax CE at catalog scale is K10 in SURVEY §2.12 (chunked
online-logsumexp HIP kernel); the sampled variant is K9 (fused
gather+dot+logsumexp).  The eager forms below are the numerics reference.
"""

from __future__ import annotations

from typing import Optional

import torch

from .base import LossBase, SampledLossBase


class CE(LossBase):
    """Full-softmax cross entropy over the catalog."""

    def forward(
        self,
        embeddings: torch.Tensor,  # [B, L, E]
        positive_labels: torch.Tensor,  # [B, L]
        padding_mask: torch.Tensor,  # [B, L] bool True=valid
        target_padding_mask: Optional[torch.Tensor] = None,
        negative_labels: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        labels = positive_labels.masked_fill(~mask, -100)
        if embeddings.is_cuda:
            from replay_amd.ops import hip_ext, require_hip_on_gpu

            head = self.logits_callback
            if (
                require_hip_on_gpu(embeddings)
                and hasattr(hip_ext(), "ce_fwd")
                and hasattr(head, "get_item_weights")
            ):
                hidden2d = embeddings.reshape(-1, embeddings.shape[-1])
                if torch.is_autocast_enabled("cuda"):
                    # custom autograd functions bypass autocast: cast here so
                    # the GEMMs and CE kernels run in bf16 (MFMA path)
                    hidden2d = hidden2d.to(torch.get_autocast_dtype("cuda"))
                labels1d = labels.reshape(-1)
                if float(mask.float().mean()) < 0.5:
                    # sparse-target objectives (BERT4Rec masks ~15% of
                    # tokens): compact to the scored rows before the CE —
                    # a 6x cut of the linear+CE work; index_select backward
                    # scatters zeros to the dropped rows
                    keep = (labels1d != -100).nonzero(as_tuple=True)[0]
                    hidden2d = hidden2d.index_select(0, keep)
                    labels1d = labels1d.index_select(0, keep)
                weight = head.get_item_weights()
                n_elems = hidden2d.shape[0] * weight.shape[0]
                if n_elems * hidden2d.element_size() > 24 * 2**30:
                    # huge catalogs (e.g. 10M items): chunked CE, logits are
                    # recomputed in backward and never fully materialized
                    from replay_amd.ops.fused_ce import chunked_fused_ce

                    return chunked_fused_ce(hidden2d, weight, labels1d, -100)
                weight = weight.to(hidden2d.dtype)
                if (
                    hidden2d.dtype == torch.bfloat16
                    and hidden2d.shape[-1] in (64, 128, 256)
                    and hasattr(hip_ext(), "ce_linear_fwd")
                ):
                    # fused linear+CE: logits never reach HBM on the forward
                    # pass; backward recomputes them and fuses dhidden
                    from replay_amd.ops.autograd import fused_linear_cross_entropy

                    return fused_linear_cross_entropy(hidden2d, weight, labels1d, -100)
                # materialized path: ONE bf16 logits buffer, fused one-pass
                # LSE forward + in-place dlogits backward (measured faster
                # than chunking at V<=1e5: no recompute, full-width GEMMs)
                from replay_amd.ops.autograd import fused_cross_entropy

                logits2d = hidden2d @ weight.t()
                return fused_cross_entropy(logits2d, labels1d, -100)
        logits = self.logits_callback(embeddings)  # [B, L, V]
        return torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(),
            labels.reshape(-1),
            ignore_index=-100,
        )


class CEWeighted(LossBase):
    """Per-position weighted full CE (reference ce.py:84)."""

    def forward(
        self,
        embeddings: torch.Tensor,
        positive_labels: torch.Tensor,
        padding_mask: torch.Tensor,
        target_padding_mask: Optional[torch.Tensor] = None,
        negative_labels: Optional[torch.Tensor] = None,
        weights: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        logits = self.logits_callback(embeddings)
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        labels = positive_labels.masked_fill(~mask, -100)
        per_pos = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]).float(),
            labels.reshape(-1),
            ignore_index=-100,
            reduction="none",
        ).reshape(labels.shape)
        if weights is None:
            weights = torch.ones_like(per_pos)
        weights = weights * mask.to(per_pos.dtype)
        return (per_pos * weights).sum() / weights.sum().clamp(min=1e-12)


class CESampled(SampledLossBase):
    """CE over [positive | negatives] sampled logits (reference ce.py:146),
    with optional uniform-sampling log-correction matching the legacy sampled
    CE (reference models/nn/sequential/sasrec/lightning.py:357-381:
    neg logits corrected by +log(V-1) - log(n_negatives))."""

    def __init__(self, log_correction: bool = False, vocab_size: Optional[int] = None) -> None:
        super().__init__()
        self.log_correction = log_correction
        self.vocab_size = vocab_size

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
            raise ValueError("CESampled requires negative_labels")
        mask = target_padding_mask if target_padding_mask is not None else padding_mask
        from replay_amd.ops.sampled_ce import can_fuse_sampled_ce, fused_sampled_ce_parts

        if can_fuse_sampled_ce(embeddings, negative_labels, self.logits_callback):
            # K9 fused path: pool LSE from the MFMA linear+LSE kernel — the
            # [B, L, n] logits never materialize
            pos_logit, lse_neg, n_coll = fused_sampled_ce_parts(
                embeddings, positive_labels, negative_labels, self.logits_callback
            )
            if self.log_correction:
                n_neg = negative_labels.shape[0]
                vocab = self.vocab_size or (int(negative_labels.max()) + 1)
                correction = torch.log(
                    torch.tensor(float(max(vocab - 1, 1)), device=lse_neg.device)
                ) - torch.log((n_neg - n_coll).clamp(min=1))
                lse_neg = lse_neg + correction
            per_pos = torch.nn.functional.softplus(lse_neg - pos_logit)
            valid = mask.to(per_pos.dtype)
            if weights is not None:
                valid = valid * weights
            return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)

        pos, neg = self.get_sampled_logits(embeddings, positive_labels, negative_labels)
        if self.log_correction:
            n_neg = neg.shape[-1]
            vocab = self.vocab_size or (int(negative_labels.max()) + 1)
            rejected = torch.isinf(neg).sum(-1, keepdim=True)
            correction = torch.log(torch.tensor(float(max(vocab - 1, 1)), device=neg.device)) - torch.log(
                (n_neg - rejected).clamp(min=1).to(neg.dtype)
            )
            neg = neg + correction
        logits = torch.cat([pos, neg], dim=-1).float()  # [B, L, 1+n]
        target = torch.zeros(logits.shape[:-1], dtype=torch.long, device=logits.device)
        per_pos = torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.shape[-1]),
            target.reshape(-1),
            reduction="none",
        ).reshape(target.shape)
        valid = mask.to(per_pos.dtype)
        if weights is not None:
            valid = valid * weights
        return (per_pos * valid).sum() / valid.sum().clamp(min=1e-12)


class CESampledWeighted(CESampled):
    """Weighted variant (reference ce.py:252): weights flow through forward."""
