"""CPU numerics for every loss in the zoo against hand-rolled references
(the reference's per-module loss unit tests, tests/nn/loss/test_loss.py in
SURVEY §4, are the model for this file)."""

import math

import pytest
import torch

from replay_amd.nn.embedding import CategoricalEmbedding
from replay_amd.nn.head import EmbeddingTyingHead
from replay_amd.nn.loss import (
    BCE,
    CE,
    BCESampled,
    CESampled,
    CESampledWeighted,
    CEWeighted,
    LogInCE,
    LogOutCE,
    ScalableCrossEntropyLoss,
)

pytestmark = pytest.mark.torch

B, L, E, V = 3, 5, 8, 20


@pytest.fixture()
def head():
    torch.manual_seed(0)
    emb = CategoricalEmbedding(V, E)
    return EmbeddingTyingHead(emb)


@pytest.fixture()
def batch():
    torch.manual_seed(1)
    emb = torch.randn(B, L, E)
    labels = torch.randint(0, V, (B, L))
    mask = torch.ones(B, L, dtype=torch.bool)
    mask[:, 0] = False  # left padding
    return emb, labels, mask


def _bind(loss, head):
    loss.set_logits_callback(head)
    return loss


class TestFullLosses:
    def test_ce_matches_functional(self, head, batch):
        emb, labels, mask = batch
        loss = _bind(CE(), head)(emb, labels, mask)
        logits = head(emb).reshape(-1, V).float()
        ref = torch.nn.functional.cross_entropy(
            logits, labels.masked_fill(~mask, -100).reshape(-1), ignore_index=-100
        )
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_ce_weighted_equals_ce_with_unit_weights(self, head, batch):
        emb, labels, mask = batch
        w = torch.ones(B, L)
        lw = _bind(CEWeighted(), head)(emb, labels, mask, weights=w)
        lc = _bind(CE(), head)(emb, labels, mask)
        torch.testing.assert_close(lw, lc, atol=1e-5, rtol=1e-5)

    def test_ce_weighted_zero_weight_removes_position(self, head, batch):
        emb, labels, mask = batch
        w = torch.ones(B, L)
        w[0, 1] = 0.0
        lw = _bind(CEWeighted(), head)(emb, labels, mask, weights=w)
        mask2 = mask.clone()
        mask2[0, 1] = False
        lc = _bind(CE(), head)(emb, labels, mask2)
        torch.testing.assert_close(lw, lc, atol=1e-5, rtol=1e-5)

    def test_bce_matches_manual(self, head, batch):
        emb, labels, mask = batch
        loss = _bind(BCE(), head)(emb, labels, mask)
        logits = head(emb).float()
        target = torch.zeros_like(logits).scatter_(-1, labels.unsqueeze(-1), 1.0)
        per = torch.nn.functional.binary_cross_entropy_with_logits(logits, target, reduction="none")
        # reference normalization: catalog-summed BCE, averaged per position
        ref = (per * mask.unsqueeze(-1)).sum() / mask.sum()
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_sce_runs_and_backprops(self, head, batch):
        emb, labels, mask = batch
        emb = emb.clone().requires_grad_(True)
        loss = _bind(ScalableCrossEntropyLoss(n_buckets=4), head)(emb, labels, mask)
        assert torch.isfinite(loss)
        loss.backward()
        assert emb.grad is not None and torch.isfinite(emb.grad).all()


class TestSampledLosses:
    def test_ce_sampled_matches_manual(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([0, 3, 7, 11])
        loss = _bind(CESampled(), head)(emb, labels, mask, negative_labels=negs)
        # manual: softmax over [pos | negs] with collisions at -inf
        weights = head.get_item_weights()
        pos_logit = (emb * weights[labels]).sum(-1, keepdim=True)
        neg_logit = emb @ weights[negs].T
        coll = negs[None, None, :] == labels.unsqueeze(-1)
        neg_logit = neg_logit.masked_fill(coll, float("-inf"))
        logits = torch.cat([pos_logit, neg_logit], -1).float()
        per = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 5), torch.zeros(B * L, dtype=torch.long), reduction="none"
        ).reshape(B, L)
        ref = (per * mask).sum() / mask.sum()
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_ce_sampled_collision_masking_excludes_positive(self, head, batch):
        emb, labels, mask = batch
        # all negatives equal to the positive: the CE target column is the
        # only finite logit -> loss exactly 0
        negs = labels.unsqueeze(-1).expand(B, L, 4)
        loss = _bind(CESampled(), head)(emb, labels, mask, negative_labels=negs)
        assert float(loss) == pytest.approx(0.0, abs=1e-6)

    def test_ce_sampled_log_correction_shifts_negatives(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([0, 3, 7, 11])
        plain = _bind(CESampled(), head)(emb, labels, mask, negative_labels=negs)
        corrected = _bind(CESampled(log_correction=True, vocab_size=V), head)(
            emb, labels, mask, negative_labels=negs
        )
        assert not torch.isclose(plain, corrected)

    def test_ce_sampled_weighted(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([0, 3, 7, 11])
        w = torch.rand(B, L) + 0.5
        lw = _bind(CESampledWeighted(), head)(emb, labels, mask, negative_labels=negs, weights=w)
        assert torch.isfinite(lw)

    def test_bce_sampled_matches_manual(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([2, 5])
        loss = _bind(BCESampled(), head)(emb, labels, mask, negative_labels=negs)
        weights = head.get_item_weights()
        pos = (emb * weights[labels]).sum(-1)
        neg = emb @ weights[negs].T
        coll = negs[None, None, :] == labels.unsqueeze(-1)
        ls = torch.nn.functional.logsigmoid
        neg_ok = ~coll
        neg_term = (ls(-neg.float()) * neg_ok).sum(-1) / neg_ok.sum(-1).clamp(min=1)
        per = -(ls(pos.float()) + neg_term)
        ref = (per * mask).sum() / mask.sum()
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_login_ce_matches_manual(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([1, 4, 9])
        t = 0.5
        loss = _bind(LogInCE(temperature=t), head)(emb, labels, mask, negative_labels=negs)
        weights = head.get_item_weights()
        pos = (emb * weights[labels]).sum(-1, keepdim=True)
        neg = emb @ weights[negs].T
        coll = negs[None, None, :] == labels.unsqueeze(-1)
        neg = neg.masked_fill(coll, float("-inf"))
        logits = torch.cat([pos, neg], -1).float() / t
        per = torch.logsumexp(logits, -1) - pos.squeeze(-1).float() / t
        ref = (per * mask).sum() / mask.sum()
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_logout_ce_positive_out_of_denominator(self, head, batch):
        emb, labels, mask = batch
        negs = torch.tensor([1, 4, 9])
        loss = _bind(LogOutCE(), head)(emb, labels, mask, negative_labels=negs)
        weights = head.get_item_weights()
        pos = (emb * weights[labels]).sum(-1).float()
        neg = (emb @ weights[negs].T).float()
        coll = negs[None, None, :] == labels.unsqueeze(-1)
        neg = neg.masked_fill(coll, float("-inf"))
        per = torch.nn.functional.softplus(torch.logsumexp(neg, -1) - pos)
        ref = (per * mask).sum() / mask.sum()
        torch.testing.assert_close(loss, ref, atol=1e-5, rtol=1e-5)

    def test_per_position_negatives_shape(self, head, batch):
        emb, labels, mask = batch
        negs = torch.randint(0, V, (B, L, 6))
        loss = _bind(CESampled(), head)(emb, labels, mask, negative_labels=negs)
        assert torch.isfinite(loss)


class TestLossCallbackBinding:
    def test_unbound_loss_raises(self, batch):
        emb, labels, mask = batch
        with pytest.raises(RuntimeError, match="logits_callback"):
            CE()(emb, labels, mask)

    def test_head_not_in_loss_state_dict(self, head):
        loss = _bind(CE(), head)
        # the bound head must NOT become a child module (duplicate params)
        assert loss.state_dict() == {}
        assert list(loss.parameters()) == []
