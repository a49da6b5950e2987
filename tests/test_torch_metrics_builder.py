import numpy as np
import pytest
import torch

from replay_amd.metrics import MAP, MRR, NDCG, HitRate, Precision, Recall, TorchMetricsBuilder

pytestmark = pytest.mark.torch


def _offline_equivalent(preds, gts, metric_cls, k):
    recs = {b: preds[b].tolist() for b in range(len(preds))}
    gt = {b: [g for g in gts[b].tolist() if g >= 0] for b in range(len(gts))}
    out = metric_cls([k])(recs, gt)
    return out[f"{metric_cls.__name__}@{k}"]


@pytest.mark.parametrize("k", [1, 3, 5])
@pytest.mark.parametrize(
    "metric_name,metric_cls",
    [
        ("recall", Recall),
        ("precision", Precision),
        ("ndcg", NDCG),
        ("map", MAP),
        ("mrr", MRR),
        ("hitrate", HitRate),
    ],
)
def test_builder_matches_offline(metric_name, metric_cls, k, rng):
    B, K, G, V = 32, 5, 4, 50
    preds = torch.stack([torch.from_numpy(rng.choice(V, size=K, replace=False)) for _ in range(B)])
    # unique ground-truth ids per user (builder contract), some rows padded
    gts = torch.stack([torch.from_numpy(rng.choice(V, size=G, replace=False)) for _ in range(B)]).long()
    pad_mask = torch.from_numpy(rng.random((B, G)) < 0.3)
    pad_mask[:, 0] = False  # ensure at least one valid
    gts[pad_mask] = -1
    builder = TorchMetricsBuilder(metrics=[metric_name], top_k=[k])
    builder.add_prediction(preds, gts)
    got = builder.get_metrics()[f"{metric_name}@{k}"]
    expected = _offline_equivalent(preds, gts, metric_cls, k)
    assert got == pytest.approx(expected, abs=1e-9)


def test_builder_batch_accumulation(rng):
    B, K, V = 16, 5, 30
    preds = torch.stack([torch.from_numpy(rng.choice(V, size=K, replace=False)) for _ in range(B)])
    gts = torch.from_numpy(rng.choice(V, size=(B, 3))).long()
    b_all = TorchMetricsBuilder(metrics=["recall"], top_k=[K])
    b_all.add_prediction(preds, gts)
    b_split = TorchMetricsBuilder(metrics=["recall"], top_k=[K])
    b_split.add_prediction(preds[:8], gts[:8])
    b_split.add_prediction(preds[8:], gts[8:])
    assert b_all.get_metrics() == pytest.approx(b_split.get_metrics())


def test_builder_coverage_and_novelty():
    preds = torch.tensor([[0, 1, 2], [3, 4, 5]])
    gts = torch.tensor([[1, -1], [9, -1]])
    train = torch.tensor([[0, 1, -1], [7, 8, 9]])
    builder = TorchMetricsBuilder(metrics=["coverage", "novelty"], top_k=[3], item_count=10)
    builder.add_prediction(preds, gts, train)
    out = builder.get_metrics()
    # train catalog = {0,1,7,8,9}; recommended = {0..5}; covered = {0,1}
    assert out["coverage@3"] == pytest.approx(2 / 5)
    # user0: 0,1 seen -> novelty 1/3; user1: none seen -> 1.0
    assert out["novelty@3"] == pytest.approx((1 / 3 + 1.0) / 2)
