"""K13 parity: the one-launch HIP metrics reduction vs the eager builder."""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")


@requires_gpu
@pytest.mark.parametrize("with_train", [False, True])
@pytest.mark.parametrize("shape", [(64, 10, 7, 30), (513, 20, 40, 120), (1000, 5, 1, 8)])
def test_metrics_kernel_matches_eager(with_train, shape):
    from replay_amd.metrics.torch_metrics_builder import TorchMetricsBuilder

    B, K, G, T = shape
    torch.manual_seed(B + K)
    V = 10_000
    preds = torch.randint(0, V, (B, K), device="cuda")
    gt = torch.randint(0, V, (B, G), device="cuda")
    gt[torch.rand(B, G, device="cuda") < 0.3] = -1
    # plant guaranteed hits in some rows
    gt[: B // 2, 0] = preds[: B // 2, min(2, K - 1)]
    train = None
    if with_train:
        train = torch.randint(0, V, (B, T), device="cuda")
        train[: B // 3, 0] = preds[: B // 3, 0]

    metrics = ["hitrate", "recall", "precision", "ndcg", "map", "mrr", "novelty", "coverage"]
    ks = [1, min(5, K), K]
    ks = sorted(set(ks))

    gpu_b = TorchMetricsBuilder(metrics, ks, item_count=V)
    assert gpu_b._try_kernel(preds[:, :K], gt, train) or pytest.fail("kernel path not taken")
    gpu_b.reset()
    gpu_b.add_prediction(preds, gt, train)
    got = gpu_b.get_metrics()

    cpu_b = TorchMetricsBuilder(metrics, ks, item_count=V)
    cpu_b.add_prediction(preds.cpu(), gt.cpu(), train.cpu() if train is not None else None)
    want = cpu_b.get_metrics()

    assert set(got) == set(want)
    for k, v in want.items():
        assert abs(got[k] - v) < 1e-5 * max(1.0, abs(v)), (k, got[k], v)
