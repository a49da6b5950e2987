"""RCCL collective helpers for cross-GPU sharing.

SURVEY §2.10: the reference's ``negatives_sharing`` (one negative set shared
batch-wide, models/nn/sequential/sasrec/lightning.py:432-438) becomes an RCCL
all-gather over xGMI.  ``gather_embeddings`` is differentiable (gradients
reduce-scatter back); ``gather_ids`` shares sampled ids (no grad).  Both are
no-ops when torch.distributed is not initialized, so single-GPU and CPU unit
tests exercise the same call sites.
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.distributed as dist


def world_info() -> Tuple[int, int]:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def gather_ids(ids: torch.Tensor) -> torch.Tensor:
    """All-gather an id tensor across ranks (uniform shape required)."""
    _, world = world_info()
    if world == 1:
        return ids
    out = [torch.empty_like(ids) for _ in range(world)]
    dist.all_gather(out, ids.contiguous())
    return torch.cat(out)


class _GatherEmbeddings(torch.autograd.Function):
    """Differentiable all-gather: forward concatenates every rank's rows;
    backward returns this rank's slice of the (summed) incoming grads via
    reduce-scatter semantics (each rank's grad slice is all-reduced)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor):
        rank, world = world_info()
        ctx.rank, ctx.world = rank, world
        ctx.rows = x.shape[0]
        if world == 1:
            return x
        out = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(out, x.contiguous())
        out[rank] = x  # keep the autograd-connected local shard
        return torch.cat(out)

    @staticmethod
    def backward(ctx, grad):
        if ctx.world == 1:
            return grad
        grad = grad.contiguous()
        dist.all_reduce(grad)
        start = ctx.rank * ctx.rows
        return grad[start : start + ctx.rows]


def gather_embeddings(x: torch.Tensor) -> torch.Tensor:
    return _GatherEmbeddings.apply(x)


def sync_sparse_grads(params) -> None:
    """Average sparse COO gradients across ranks.

    RCCL/NCCL all-reduce requires dense tensors, so sparse embedding grads
    (K6: touched-rows-only at 10M+ catalogs) are synchronized by an explicit
    all-gather of (indices, values) padded to the max nnz, then coalesced.
    Payload is O(touched rows x E) — tiny next to the dense [V, E] all-reduce
    it replaces.
    """
    rank, world = world_info()
    if world == 1:
        return
    for p in params:
        g = p.grad
        if g is None or not g.is_sparse:
            continue
        g = g.coalesce()
        idx, val = g.indices(), g.values()
        n = torch.tensor([idx.shape[1]], device=val.device, dtype=torch.long)
        counts = [torch.zeros_like(n) for _ in range(world)]
        dist.all_gather(counts, n)
        maxn = int(torch.stack(counts).max())
        pad_idx = torch.zeros(idx.shape[0], maxn, dtype=idx.dtype, device=idx.device)
        pad_val = torch.zeros((maxn,) + tuple(val.shape[1:]), dtype=val.dtype, device=val.device)
        pad_idx[:, : idx.shape[1]] = idx
        pad_val[: val.shape[0]] = val
        gi = [torch.empty_like(pad_idx) for _ in range(world)]
        gv = [torch.empty_like(pad_val) for _ in range(world)]
        dist.all_gather(gi, pad_idx.contiguous())
        dist.all_gather(gv, pad_val.contiguous())
        cat_i = torch.cat([t[:, : int(c)] for t, c in zip(gi, counts)], dim=1)
        cat_v = torch.cat([t[: int(c)] for t, c in zip(gv, counts)], dim=0)
        p.grad = torch.sparse_coo_tensor(cat_i, cat_v / world, g.shape).coalesce()
