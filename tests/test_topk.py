"""catalog_topk correctness vs exhaustive reference (K7/K8)."""

import pytest
import torch

from replay_amd.ops.topk import catalog_topk

pytestmark = pytest.mark.torch


@pytest.mark.parametrize("chunk", [7, 64, 1000])
def test_catalog_topk_matches_full(chunk):
    torch.manual_seed(0)
    B, E, V, K = 8, 16, 100, 5
    q = torch.randn(B, E)
    items = torch.randn(V, E)
    scores, ids = catalog_topk(q, items, K, chunk_items=chunk)
    full = q @ items.T
    ref_s, ref_i = torch.topk(full, K, dim=1)
    torch.testing.assert_close(scores, ref_s)
    assert ids.tolist() == ref_i.tolist()


def test_catalog_topk_filter_seen():
    torch.manual_seed(1)
    B, E, V, K = 4, 8, 50, 3
    q = torch.randn(B, E)
    items = torch.randn(V, E)
    full = q @ items.T
    ref_i = torch.topk(full, 1, dim=1).indices  # mask each query's best item
    seen = torch.full((B, 2), -1, dtype=torch.long)
    seen[:, 0] = ref_i[:, 0]
    scores, ids = catalog_topk(q, items, K, seen=seen, chunk_items=16)
    for b in range(B):
        assert int(ref_i[b, 0]) not in ids[b].tolist()
    # equals exhaustive with the same mask
    masked = full.clone()
    masked[torch.arange(B), ref_i[:, 0]] = float("-inf")
    ref2 = torch.topk(masked, K, dim=1)
    assert ids.tolist() == ref2.indices.tolist()


def test_sharded_matches_single_process_fallback():
    from replay_amd.ops.topk import sharded_catalog_topk

    torch.manual_seed(2)
    q = torch.randn(4, 8)
    items = torch.randn(40, 8)
    s1, i1 = sharded_catalog_topk(q, items, 5, shard_offset=0)
    s2, i2 = catalog_topk(q, items, 5)
    assert i1.tolist() == i2.tolist()
