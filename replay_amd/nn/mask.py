"""Attention-mask construction.

Parity with reference replay/nn/mask.py (DefaultAttentionMask:58, float mask
build :30-51): causal tril AND key-padding, OR'd with the diagonal so a fully
masked row still attends to itself; fill value -inf in train, finfo.min in
eval (reference :39-42).

MI355X design note: the float [B*H, L, L] mask is built LAZILY via
:class:`MaskSpec` — on the GPU path the fused attention kernel (K1/K2)
consumes the [B, L] bool padding mask + causal flag directly and the float
tensor never materializes.
"""

from __future__ import annotations

import torch


class MaskSpec:
    """Lazy attention-mask: holds what the fused kernel needs; materializes
    the additive float mask only for eager consumers."""

    def __init__(self, padding_mask: torch.Tensor, num_heads: int, causal: bool, training: bool) -> None:
        self.padding_mask = padding_mask  # [B, L] bool, True = valid
        self.num_heads = num_heads
        self.causal = causal
        self.training = training

    def materialize(self) -> torch.Tensor:
        B, L = self.padding_mask.shape
        device = self.padding_mask.device
        allowed = self.padding_mask[:, None, :].expand(B, L, L)
        if self.causal:
            causal = torch.tril(torch.ones(L, L, dtype=torch.bool, device=device))
            allowed = allowed & causal[None]
        diag = torch.eye(L, dtype=torch.bool, device=device)[None].expand(B, L, L)
        allowed = allowed | diag
        fill = float("-inf") if self.training else torch.finfo(torch.float32).min
        mask = torch.zeros(B, L, L, dtype=torch.float32, device=device)
        mask = mask.masked_fill(~allowed, fill)
        if self.num_heads > 1:
            mask = mask.repeat_interleave(self.num_heads, dim=0)
        return mask


class DefaultAttentionMask(torch.nn.Module):
    def __init__(self, num_heads: int = 1, causal: bool = True) -> None:
        super().__init__()
        self.num_heads = num_heads
        self.causal = causal

    def forward(self, padding_mask: torch.Tensor) -> MaskSpec:
        """padding_mask: [B, L] bool (True = valid).  Returns a MaskSpec;
        call ``.materialize()`` for the additive float [B*H, L, L] mask."""
        return MaskSpec(padding_mask, self.num_heads, self.causal, self.training)
