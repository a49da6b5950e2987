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


class FusedCrossEntropyFunction(torch.autograd.Function):
    """Fused softmax-CE over [N, V] logits (K10): forward reads the logits
    twice (max, sum-exp) with no fp32 copy; backward overwrites the logits
    storage with dlogits in one pass."""

    @staticmethod
    def forward(ctx, logits2d: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100):
        ext = hip_ext()
        loss_sum, count, lse = ext.ce_fwd(logits2d, labels, ignore_index)
        ctx.save_for_backward(logits2d, labels, lse, count)
        ctx.ignore_index = ignore_index
        return (loss_sum / count.clamp(min=1).to(loss_sum.dtype)).squeeze(0)

    @staticmethod
    def backward(ctx, dloss):
        ext = hip_ext()
        logits2d, labels, lse, count = ctx.saved_tensors
        dlogits = ext.ce_bwd(
            logits2d, labels, lse, dloss.reshape(1), count, ctx.ignore_index, True
        )
        return dlogits, None, None


def fused_cross_entropy(logits2d: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100):
    return FusedCrossEntropyFunction.apply(logits2d, labels, ignore_index)


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
