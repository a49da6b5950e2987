"""Attention-mask construction.

Parity with reference replay/nn/mask.py (DefaultAttentionMask:58, float mask
build :30-51): causal tril AND key-padding, OR'd with the diagonal so a fully
masked row still attends to itself; fill value -inf in train, finfo.min in
eval (reference :39-42).
"""

from __future__ import annotations

import torch


class DefaultAttentionMask(torch.nn.Module):
    def __init__(self, num_heads: int = 1, causal: bool = True) -> None:
        super().__init__()
        self.num_heads = num_heads
        self.causal = causal

    def forward(self, padding_mask: torch.Tensor) -> torch.Tensor:
        """padding_mask: [B, L] bool (True = valid).  Returns float mask
        [B*H, L, L] additive."""
        B, L = padding_mask.shape
        device = padding_mask.device
        allowed = padding_mask[:, None, :].expand(B, L, L)  # key validity
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
