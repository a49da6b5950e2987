"""torch.autograd.Function wrappers around the HIP kernels."""

from __future__ import annotations

import torch

from replay_amd.ops import hip_ext


class LayerNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = hip_ext()
        x = x.contiguous()
        y, mean, rstd = ext.layer_norm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = hip_ext()
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.layer_norm_bwd(x, dy.contiguous(), weight, mean, rstd)
        return dx, dw.to(weight.dtype), db.to(weight.dtype), None


class FlashAttentionFunction(torch.autograd.Function):
    """Placeholder until the HIP flash kernel lands; the dispatch in
    ops/attention.py only routes here when the extension exports
    attention_fwd."""

    @staticmethod
    def forward(ctx, q, k, v, attn_mask):  # pragma: no cover
        raise NotImplementedError

    @staticmethod
    def backward(ctx, do):  # pragma: no cover
        raise NotImplementedError
