"""Linear with a chunked-reduction weight gradient (MI355X-tuned).

The transformer projections of the flagship shapes are skinny GEMMs with a
huge reduction dimension (dW = dy^T x with [N=409600] rows and 64-384
outputs).  hipBLASLt/rocBLAS have no split-K solution for that `nt` shape
on gfx950 — the best tuned pick runs ~340-690 us where the HBM read floor
is ~35 us (measured, tools/wgrad_orientations.py).  Reformulating the
reduction as a chunked batched GEMM + partial-sum
(``(dy.view(C,N/C,F).mT @ x.view(C,N/C,E)).sum(0)``) reaches 36-81 us —
the bmm solution space tiles the reduction across workgroups, which is
exactly what split-K would have done.

``ChunkedWgradLinear`` is a drop-in ``torch.nn.Linear`` subclass (same
parameters / state-dict layout); only the backward weight-gradient path
differs.  Chunk partials accumulate in fp32 inside the MFMA GEMM; the
final ``.sum(0)`` also runs in fp32, so the only extra rounding vs a
monolithic GEMM is the C bf16 partial stores — relative error ~4e-3 /
sqrt(C), far inside training-grad noise (parity-tested).

Reference surface: the projections inside
``replay/nn/sequential/sasrec/transformer.py:93-110`` and
``replay/nn/ffn.py:11-133`` (torch.nn.MultiheadAttention /
conv1d-as-GEMM) — behavior identical, only the backward schedule is
MI355X-specific.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

# reduction chunking kicks in above this many rows (below it, the plain
# GEMM is already at the floor and the extra partial-sum launch loses)
_MIN_ROWS = 65536


def _chunked_wgrad(dy2d: torch.Tensor, x2d: torch.Tensor) -> torch.Tensor:
    n = dy2d.shape[0]
    c = 64
    while c > 1 and n % c:
        c //= 2
    if c == 1:
        return dy2d.t() @ x2d
    dyc = dy2d.view(c, n // c, dy2d.shape[1])
    xc = x2d.view(c, n // c, x2d.shape[1])
    return (dyc.transpose(1, 2) @ xc).float().sum(0)


class _ChunkedWgradLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor]):
        ctx.save_for_backward(x2d, weight)
        ctx.has_bias = bias is not None
        return F.linear(x2d, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dy @ weight
        if dy.shape[0] >= _MIN_ROWS:
            dw = _chunked_wgrad(dy, x2d)
        else:
            dw = dy.t() @ x2d
        db = dy.sum(0) if ctx.has_bias else None
        return dx, dw.to(weight.dtype), db


class ChunkedWgradLinear(torch.nn.Linear):
    """torch.nn.Linear whose weight gradient uses the chunked reduction."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # noqa: D102
        w, b = self.weight, self.bias
        if torch.is_autocast_enabled() and x.is_cuda:
            dt = torch.get_autocast_gpu_dtype()
            x = x.to(dt)
            w = w.to(dt)
            b = b.to(dt) if b is not None else None
        lead = x.shape[:-1]
        y = _ChunkedWgradLinearFn.apply(x.reshape(-1, x.shape[-1]), w, b)
        return y.view(*lead, y.shape[-1])
