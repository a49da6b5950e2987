"""Catalog-scale top-K serving (BASELINE config 5 shape, CPU-sized here):
score a query embedding against the full item table and return the top-K
unseen items.  On an MI355X with k <= 512 this dispatches to the fused MFMA
score-GEMM whose epilogue selects candidates before the [B, V] score matrix
ever reaches HBM (replay_amd/ops/hip/scored_topk_gemm.hip); at 10M items /
d=256 it serves 57K queries/s on one GPU (profiles/PROFILES.md).  Across
GPUs the catalog is sharded and the per-shard top-K merged with one small
all-gather (sharded_catalog_topk)."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))  # repo root

import torch

from replay_amd.ops.topk import catalog_topk

B, V, E, K = 64, 50_000, 64, 10


def main():
    torch.manual_seed(0)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    item_table = torch.randn(V, E, device=device, dtype=dtype)
    queries = torch.randn(B, E, device=device, dtype=dtype)

    # per-query seen history (global item ids, -1 padded) — filtered exactly
    seen = torch.randint(0, V, (B, 30), device=device)

    scores, ids = catalog_topk(queries, item_table, K, seen=seen)
    print("scores", scores.shape, "ids", ids.shape)

    # contract checks: K per row, no seen item recommended
    assert ids.shape == (B, K)
    for b in range(B):
        assert not set(ids[b].tolist()) & set(seen[b].tolist())
    print("top-10 for first 3 queries:\n", ids[:3])


if __name__ == "__main__":
    main()
