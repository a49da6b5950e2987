"""Fused attention dispatch: scores + mask + softmax + PV.

CPU path: explicit matmul/softmax eager math (the numerics reference).
GPU path: the gfx950 flash-style HIP kernel (K1/K2 in SURVEY §2.12) via the
in-tree extension; falls back loudly per replay_amd.ops.require_hip_on_gpu.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from replay_amd.ops import require_hip_on_gpu


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
    """q,k,v: [BH, L, Dh]; attn_mask additive [BH, L, L] or None.

    The flash-style HIP kernel takes a [B,L] bool padding mask + causal flag,
    not an additive [BH,L,L] mask; callers with a MaskSpec dispatch to it
    directly in ``MultiheadAttention.forward``.  Additive-mask callers land
    here and run the eager math — on GPU that is still torch-ROCm
    (hipBLASLt GEMMs + fused softmax), the correct path for materialized
    masks, odd dtypes and dropout.
    """
    require_hip_on_gpu(q)  # loud failure if the extension is missing on GPU
    return eager_attention(q, k, v, attn_mask, dropout_p)
