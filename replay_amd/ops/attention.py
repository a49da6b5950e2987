"""Fused attention dispatch: scores + mask + softmax + PV.

CPU path: explicit matmul/softmax eager math (the numerics reference).
GPU path: the gfx950 flash-style HIP kernel (K1/K2 in SURVEY §2.12) via the
in-tree extension; falls back loudly per replay_amd.ops.require_hip_on_gpu.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from replay_amd.ops import hip_ext, require_hip_on_gpu


def eager_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    attn_mask: Optional[torch.Tensor],
    dropout_p: float = 0.0,
) -> torch.Tensor:
    scale = 1.0 / math.sqrt(q.shape[-1])
    scores = (q @ k.transpose(-1, -2)) * scale
    if attn_mask is not None:
        scores = scores + attn_mask
    probs = torch.softmax(scores.float(), dim=-1).to(q.dtype)
    probs = torch.nan_to_num(probs, nan=0.0)
    if dropout_p > 0.0:
        probs = torch.nn.functional.dropout(probs, dropout_p)
    return probs @ v


def fused_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    attn_mask: Optional[torch.Tensor],
    dropout_p: float = 0.0,
) -> torch.Tensor:
    """q,k,v: [BH, L, Dh]; attn_mask additive [BH, L, L] or None."""
    if require_hip_on_gpu(q):
        ext = hip_ext()
        if hasattr(ext, "attention_fwd") and dropout_p == 0.0 and q.dtype in (torch.bfloat16, torch.float16):
            from replay_amd.ops.autograd import FlashAttentionFunction

            return FlashAttentionFunction.apply(q, k, v, attn_mask)
        # HIP extension present but this config unsupported -> eager on GPU is
        # still torch-ROCm (hipBLASLt GEMMs), allowed for odd dtypes/dropout.
    return eager_attention(q, k, v, attn_mask, dropout_p)
