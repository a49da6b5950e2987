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
    storage with dlogits in one pass (halves peak memory at large V).

    The in-place consumption means the logits buffer is CORRUPTED after the
    first backward: a second backward through this node (retain_graph) is
    guarded with a RuntimeError, and when the backward itself is recorded
    (create_graph / double backward) the kernel falls back to an out-of-place
    pass on a clone so the saved logits stay intact."""

    @staticmethod
    def forward(ctx, logits2d: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100):
        ext = hip_ext()
        loss_sum, count, lse = ext.ce_fwd(logits2d, labels, ignore_index)
        ctx.save_for_backward(logits2d, labels, lse, count)
        ctx.ignore_index = ignore_index
        ctx.consumed = False
        return (loss_sum / count.clamp(min=1).to(loss_sum.dtype)).squeeze(0)

    @staticmethod
    def backward(ctx, dloss):
        ext = hip_ext()
        logits2d, labels, lse, count = ctx.saved_tensors
        if ctx.consumed:
            raise RuntimeError(
                "FusedCrossEntropyFunction.backward consumed the saved logits "
                "in-place on a previous call; re-run the forward instead of "
                "backpropagating twice through the same fused-CE node."
            )
        in_place = not torch.is_grad_enabled()
        if in_place:
            ctx.consumed = True
        else:
            logits2d = logits2d.clone()
        dlogits = ext.ce_bwd(
            logits2d, labels, lse, dloss.reshape(1), count, ctx.ignore_index, True
        )
        return dlogits, None, None


def fused_cross_entropy(logits2d: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100):
    return FusedCrossEntropyFunction.apply(logits2d, labels, ignore_index)


class FusedLinearCrossEntropyFunction(torch.autograd.Function):
    """Linear + softmax-CE without materializing [N, V] logits on the forward
    pass (ce_linear.hip): forward computes per-row LSE and the label logit in
    the MFMA GEMM epilogue; backward recomputes dlogits tile-wise, fuses the
    dhidden GEMM in-kernel (E <= 128), and materializes dlogits once for the
    weight-gradient hipBLASLt GEMM."""

    @staticmethod
    def forward(ctx, hidden2d: torch.Tensor, weight: torch.Tensor, labels: torch.Tensor,
                ignore_index: int = -100):
        ext = hip_ext()
        hidden2d = hidden2d.contiguous()
        weight = weight.contiguous()
        if hasattr(ext, "ce_linear_lse"):
            # LSE-only kernel (the in-kernel per-logit label compare was ~2%
            # VALU issue); the label logit is a gather + row dot here
            lse = ext.ce_linear_lse(hidden2d, weight)
            lab_w = weight.index_select(0, labels.clamp(min=0))
            lab_logit = (hidden2d.float() * lab_w.float()).sum(-1)
        else:
            lse, lab_logit = ext.ce_linear_fwd(hidden2d, weight, labels)
        valid = labels != ignore_index
        count = valid.sum()
        loss = torch.where(valid, lse - lab_logit, torch.zeros_like(lse)).sum()
        loss = loss / count.clamp(min=1).to(loss.dtype)
        ctx.save_for_backward(hidden2d, weight, labels, lse, valid, count)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        import os

        ext = hip_ext()
        hidden2d, weight, labels, lse, valid, count = ctx.saved_tensors
        g = dloss.to(torch.float32) / count.clamp(min=1).to(torch.float32)
        # the kernel folds |g| into adj = lse - ln|g|; the (uniform) sign of
        # the upstream gradient travels as a scalar
        gsign = float(torch.sign(g))
        gscale = torch.where(valid, g.abs(), torch.zeros((), device=valid.device))
        if (
            hidden2d.shape[1] <= 128
            and hasattr(ext, "ce_linear_wgrad")
            and os.environ.get("REPLAY_AMD_CE_WGRAD") == "1"
        ):
            # OPT-IN phase-split backward (REPLAY_AMD_CE_WGRAD=1): dW from the
            # item-owner wgrad kernel and dhidden from the store-free fused-dh
            # pass — the [M, Vp] bf16 dlogits tensor never exists, but both
            # kernels are LDS-transpose-stage bound and MEASURED SLOWER than
            # the shipped dlogits+hipBLASLt path (11.6 + 11.5 ms vs ~12 ms
            # total at the flagship shape, i.e. the memory saving costs ~7 ms
            # of step time).  Kept for the T10 (ds_read_b64_tr_b16 hardware
            # transpose) rework; parity-tested either way.
            dweight = ext.ce_linear_wgrad(hidden2d, weight, labels, lse, gscale, gsign)
            dhidden = ext.ce_linear_bwd_fused_dh(hidden2d, weight, labels, lse, gscale, gsign)
            return dhidden, dweight.to(weight.dtype), None, None
        dlogits, dhidden = ext.ce_linear_bwd(hidden2d, weight, labels, lse, gscale, gsign)
        if dhidden.numel() == 0:  # E > 128: host GEMM for the input gradient
            dhidden = dlogits @ weight
        dweight = dlogits.t() @ hidden2d
        return dhidden, dweight.to(weight.dtype), None, None


def fused_linear_cross_entropy(hidden2d: torch.Tensor, weight: torch.Tensor,
                               labels: torch.Tensor, ignore_index: int = -100):
    return FusedLinearCrossEntropyFunction.apply(hidden2d, weight, labels, ignore_index)


class FlashAttentionFunction(torch.autograd.Function):
    """Fused attention (K1/K2): q,k,v [B,H,L,Dh]; padding_mask [B,L] bool
    (True = valid); causal flag.  Backward recomputes P from the saved LSE."""

    @staticmethod
    def forward(ctx, q, k, v, padding_mask, causal):
        ext = hip_ext()
        scale = 1.0 / (q.shape[-1] ** 0.5)
        need_lse = q.requires_grad or k.requires_grad or v.requires_grad
        if (
            q.dtype == torch.bfloat16
            and q.shape[-1] in (32, 64)
            and q.shape[2] <= 256
            and hasattr(ext, "attention_fwd_mfma")
        ):
            out, lse = ext.attention_fwd_mfma(q, k, v, padding_mask, scale, causal, need_lse)
        else:
            out, lse = ext.attention_fwd(q, k, v, padding_mask, scale, causal, need_lse)
        ctx.save_for_backward(q, k, v, out, lse, padding_mask)
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = hip_ext()
        q, k, v, out, lse, padding_mask = ctx.saved_tensors
        if (
            q.dtype == torch.bfloat16
            and q.shape[-1] in (32, 64)
            and q.shape[2] <= 256
            and hasattr(ext, "attention_bwd_mfma")
        ):
            dq, dk, dv = ext.attention_bwd_mfma(
                q, k, v, out, dout, lse, padding_mask, ctx.scale, ctx.causal
            )
        else:
            dq, dk, dv = ext.attention_bwd(
                q, k, v, out, dout, lse, padding_mask, ctx.scale, ctx.causal
            )
        return dq, dk, dv, None, None
