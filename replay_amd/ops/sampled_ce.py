"""Fused sampled-softmax CE over a shared negative pool (K9 in SURVEY §2.12).

The eager path materializes neg_logits [B, L, n] (6.7 GB bf16 at the
config-4 shape B=8192, L=50, n=8192) just to reduce it to a per-position
logsumexp.  Here the pool LSE comes out of the fused linear+LSE MFMA kernel
(ce_linear.hip) with the pool table as the "catalog" — the [N, n] logits
never exist in HBM on the forward pass; the backward recomputes them in
row chunks through hipBLASLt (bounded workspace), exactly the ce_linear
recompute strategy.

Collision handling (reference replay/nn/loss/base.py:157 masks negatives
equal to the position's positive to -inf) uses the identity that a
colliding pool column's logit IS the positive logit:

    lse_excl = lse_all + log1p(-n_coll * exp(pos - lse_all))

so the fused kernel needs no per-row masking; ``n_coll`` comes from a
searchsorted count over the sorted pool (no [N, n] compare).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from replay_amd.ops import hip_ext


class NegPoolLSE(torch.autograd.Function):
    """lse over h2d @ wneg^T per row: fused MFMA forward, chunked hipBLASLt
    recompute backward."""

    CHUNK = 65536

    @staticmethod
    def forward(ctx, h2d: torch.Tensor, wneg: torch.Tensor) -> torch.Tensor:
        ext = hip_ext()
        h_b = h2d.to(torch.bfloat16).contiguous()
        w_b = wneg.to(torch.bfloat16).contiguous()
        if hasattr(ext, "ce_linear_lse"):
            lse = ext.ce_linear_lse(h_b, w_b)
        else:
            zeros = torch.zeros(h_b.shape[0], dtype=torch.long, device=h_b.device)
            lse, _ = ext.ce_linear_fwd(h_b, w_b, zeros)
        ctx.save_for_backward(h_b, w_b, lse)
        ctx.h_dtype = h2d.dtype
        ctx.w_dtype = wneg.dtype
        return lse  # [N] fp32

    @staticmethod
    def backward(ctx, dlse: torch.Tensor):
        h_b, w_b, lse = ctx.saved_tensors
        N = h_b.shape[0]
        dh = torch.empty_like(h_b, dtype=torch.float32)
        dwneg = torch.zeros_like(w_b, dtype=torch.float32)
        dlse = dlse.float()
        for lo in range(0, N, NegPoolLSE.CHUNK):
            hi = min(lo + NegPoolLSE.CHUNK, N)
            hc = h_b[lo:hi]
            logits = (hc @ w_b.T).float()  # [C, n], bf16 GEMM + fp32 read
            p = torch.exp(logits - lse[lo:hi, None]) * dlse[lo:hi, None]
            pb = p.to(torch.bfloat16)
            dh[lo:hi] = (pb @ w_b).float()
            dwneg += (pb.T @ hc).float()
        return dh.to(ctx.h_dtype), dwneg.to(ctx.w_dtype)


def neg_pool_lse(h2d: torch.Tensor, wneg: torch.Tensor) -> torch.Tensor:
    return NegPoolLSE.apply(h2d, wneg)


def can_fuse_sampled_ce(embeddings: torch.Tensor, negative_labels: torch.Tensor, callback) -> bool:
    import os

    if os.environ.get("REPLAY_AMD_NO_FUSED_SCE") == "1":
        return False
    ext = hip_ext()
    return (
        embeddings.is_cuda
        and ext is not None
        and (hasattr(ext, "ce_linear_lse") or hasattr(ext, "ce_linear_fwd"))
        and negative_labels.dim() == 1
        and negative_labels.shape[0] >= 512
        and embeddings.shape[-1] in (64, 128, 256)
        and hasattr(callback, "get_item_weights")
    )


def fused_sampled_ce_parts(
    embeddings: torch.Tensor,  # [B, L, E] (or [N, E])
    positive_labels: torch.Tensor,  # [B, L]
    negative_labels: torch.Tensor,  # [n] shared pool
    callback,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (pos_logit [B, L], lse_excl [B, L], n_coll [B, L]) — the
    ingredients of any sampled loss over a shared pool, computed without
    materializing [B, L, n]."""
    shape = positive_labels.shape
    E = embeddings.shape[-1]
    h2d = embeddings.reshape(-1, E)
    pos = positive_labels.clamp(min=0).reshape(-1)

    wneg = callback.get_item_weights(negative_labels)  # [n, E] (differentiable gather)
    pos_w = callback.get_item_weights(pos)  # [N, E]
    pos_logit = (h2d.float() * pos_w.float()).sum(-1)  # [N]

    lse_all = neg_pool_lse(h2d, wneg)  # [N] fp32

    with torch.no_grad():
        sorted_pool, _ = negative_labels.sort()
        left = torch.searchsorted(sorted_pool, pos, right=False)
        right = torch.searchsorted(sorted_pool, pos, right=True)
        n_coll = (right - left).to(torch.float32)  # [N]

    # exclude colliding columns: their logit equals pos_logit exactly (same
    # table row), so subtract their mass in log space
    x = (n_coll * torch.exp(pos_logit - lse_all)).clamp(max=1.0 - 1e-6)
    lse_excl = lse_all + torch.log1p(-x)
    return (
        pos_logit.reshape(shape),
        lse_excl.reshape(shape),
        n_coll.reshape(shape),
    )
