"""GPU parity: fused sampled-CE (K9) vs the eager CESampled.

The fused path computes the shared-pool LSE with the ce_linear MFMA kernel
(no [B, L, n] logits); it must match the eager gather+cat+cross_entropy
numerics (reference replay/models/nn/sequential/sasrec/lightning.py:357-381
semantics incl. log-correction and collision rejection) in value AND in the
gradients that reach the hidden states and the item table.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")


def _setup(B, L, E, V, n_neg, seed=0, force_collisions=True):
    from replay_amd.nn.embedding import CategoricalEmbedding
    from replay_amd.nn.head import EmbeddingTyingHead

    torch.manual_seed(seed)
    emb = CategoricalEmbedding(V, E).cuda()
    head = EmbeddingTyingHead(emb)
    h = torch.randn(B, L, E, device="cuda", requires_grad=True)
    labels = torch.randint(0, V, (B, L), device="cuda")
    negs = torch.randint(0, V, (n_neg,), device="cuda")
    if force_collisions:  # guarantee some collision rows
        negs[: B] = labels[:, 0]
    mask = torch.rand(B, L, device="cuda") > 0.1
    mask[:, 0] = True
    return head, emb, h, labels, negs, mask


@requires_gpu
class TestFusedSampledCE:
    @pytest.mark.parametrize("log_correction", [False, True])
    @pytest.mark.parametrize("E", [64, 128])
    def test_loss_and_grad_parity(self, log_correction, E):
        from replay_amd.nn.loss import CESampled
        from replay_amd.ops.sampled_ce import can_fuse_sampled_ce

        B, L, V, n_neg = 16, 12, 5000, 2048
        head, emb, h, labels, negs, mask = _setup(B, L, E, V, n_neg)

        loss_fused = CESampled(log_correction=log_correction, vocab_size=V)
        loss_fused.set_logits_callback(head)
        assert can_fuse_sampled_ce(h, negs, head)
        out_f = loss_fused(h, labels, mask, negative_labels=negs)
        gh_f, gw_f = torch.autograd.grad(out_f, [h, emb.item_emb.weight], retain_graph=False)

        # eager reference: force the fallback by a CPU-side monkeypatch
        import replay_amd.ops.sampled_ce as sce

        orig = sce.can_fuse_sampled_ce
        sce.can_fuse_sampled_ce = lambda *a, **k: False
        try:
            loss_e = CESampled(log_correction=log_correction, vocab_size=V)
            loss_e.set_logits_callback(head)
            out_e = loss_e(h, labels, mask, negative_labels=negs)
            gh_e, gw_e = torch.autograd.grad(out_e, [h, emb.item_emb.weight])
        finally:
            sce.can_fuse_sampled_ce = orig

        assert torch.isfinite(out_f)
        # bf16 kernel vs fp32 eager: value tolerance scales with logit spread
        assert abs(float(out_f) - float(out_e)) < 0.02 * max(1.0, abs(float(out_e)))
        cos_h = torch.nn.functional.cosine_similarity(gh_f.flatten(), gh_e.flatten(), dim=0)
        cos_w = torch.nn.functional.cosine_similarity(gw_f.flatten(), gw_e.flatten(), dim=0)
        assert float(cos_h) > 0.999, float(cos_h)
        assert float(cos_w) > 0.999, float(cos_w)
        rel_h = (gh_f - gh_e).norm() / gh_e.norm().clamp(min=1e-12)
        rel_w = (gw_f - gw_e).norm() / gw_e.norm().clamp(min=1e-12)
        assert float(rel_h) < 0.05, float(rel_h)
        assert float(rel_w) < 0.05, float(rel_w)

    def test_collision_rows_match_eager(self):
        """Rows whose positive floods the pool still agree (log1p exclusion)."""
        from replay_amd.nn.loss import CESampled

        B, L, E, V, n_neg = 4, 3, 64, 100, 1024
        head, emb, h, labels, negs, mask = _setup(B, L, E, V, n_neg, seed=3)
        # heavy collisions: half the pool equals row 0's positive
        negs[:512] = labels[0, 0]
        loss = CESampled(log_correction=True, vocab_size=V)
        loss.set_logits_callback(head)
        out_f = loss(h, labels, mask, negative_labels=negs)

        import replay_amd.ops.sampled_ce as sce

        orig = sce.can_fuse_sampled_ce
        sce.can_fuse_sampled_ce = lambda *a, **k: False
        try:
            out_e = loss(h, labels, mask, negative_labels=negs)
        finally:
            sce.can_fuse_sampled_ce = orig
        assert abs(float(out_f) - float(out_e)) < 0.05 * max(1.0, abs(float(out_e)))


@requires_gpu
def test_logince_fused_parity():
    """TwoTower's InfoNCE (positives in-logits) through the fused pool-LSE."""
    from replay_amd.nn.loss import LogInCE

    B, L, E, V, n_neg = 16, 1, 128, 5000, 2048
    head, emb, h, labels, negs, mask = _setup(B, L, E, V, n_neg, seed=5)
    loss = LogInCE()
    loss.set_logits_callback(head)
    out_f = loss(h, labels, mask, negative_labels=negs)
    gh_f, gw_f = torch.autograd.grad(out_f, [h, emb.item_emb.weight])

    import replay_amd.ops.sampled_ce as sce

    orig = sce.can_fuse_sampled_ce
    sce.can_fuse_sampled_ce = lambda *a, **k: False
    try:
        out_e = loss(h, labels, mask, negative_labels=negs)
        gh_e, gw_e = torch.autograd.grad(out_e, [h, emb.item_emb.weight])
    finally:
        sce.can_fuse_sampled_ce = orig
    assert abs(float(out_f) - float(out_e)) < 0.02 * max(1.0, abs(float(out_e)))
    assert float(torch.nn.functional.cosine_similarity(gh_f.flatten(), gh_e.flatten(), dim=0)) > 0.999
    assert float(torch.nn.functional.cosine_similarity(gw_f.flatten(), gw_e.flatten(), dim=0)) > 0.999
