"""Legacy-surface NN model tests (reference layer 8 parity)."""

import pytest
import torch

from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
from replay_amd.data.schema import FeatureHint, FeatureType

pytestmark = pytest.mark.torch

V, L = 40, 10


@pytest.fixture()  # function scope: resize test mutates cardinality
def schema():
    return TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=V, embedding_dim=16,
            )
        ]
    )


@pytest.fixture()
def batch():
    torch.manual_seed(0)
    b = {
        "item_id": torch.randint(0, V, (4, L)),
        "labels": torch.randint(0, V, (4, L)),
        "padding_mask": torch.ones(4, L, dtype=torch.bool),
    }
    b["labels_padding_mask"] = b["padding_mask"]
    return b


@pytest.mark.parametrize(
    "loss_type,n_samples",
    [("CE", None), ("CE", 8), ("BCE", None), ("BCE", 8), ("SCE", None)],
)
def test_legacy_sasrec_losses(schema, batch, loss_type, n_samples):
    from replay_amd.models.nn import SasRec

    model = SasRec(schema, max_seq_len=L, hidden_size=16, block_count=1,
                   loss_type=loss_type, loss_sample_count=n_samples)
    loss = model.training_step(batch)
    loss.backward()
    assert torch.isfinite(loss)


@pytest.mark.parametrize("strategy", ["global_uniform", "inbatch"])
def test_legacy_sasrec_sampling_strategies(schema, batch, strategy):
    from replay_amd.models.nn import SasRec

    model = SasRec(schema, max_seq_len=L, hidden_size=16, block_count=1,
                   loss_type="CE", loss_sample_count=8, negative_sampling_strategy=strategy)
    loss = model.training_step(batch)
    assert torch.isfinite(loss)


def test_embedding_resize(schema, batch):
    from replay_amd.models.nn import SasRec

    model = SasRec(schema, max_seq_len=L, hidden_size=16, block_count=1)
    old = model.get_all_embeddings()["item_embedding"]
    model.set_item_embeddings_by_size(V + 5)
    new = model.get_all_embeddings()["item_embedding"]
    assert new.shape[0] == V + 5
    torch.testing.assert_close(new[:V], old)
    model.append_item_embeddings(torch.randn(3, 16))
    assert model.get_all_embeddings()["item_embedding"].shape[0] == V + 8
    # model still runs after resize
    model._model.eval()
    logits = model._model.forward_inference(batch)
    assert logits.shape == (4, V + 8)


def test_legacy_bert4rec(schema, batch):
    from replay_amd.models.nn import Bert4Rec

    model = Bert4Rec(schema, max_seq_len=L, hidden_size=16, block_count=1, head_count=2)
    loss = model.training_step(batch)
    loss.backward()
    assert torch.isfinite(loss)


def test_tisasrec(schema, batch):
    from replay_amd.models.nn import TiSasRec

    model = TiSasRec(schema, max_sequence_length=L, embedding_dim=16, num_blocks=1, dropout=0.0)
    b = dict(batch)
    b["timestamp"] = torch.cumsum(torch.randint(1, 20, (4, L)), dim=1)
    loss = model(b)
    loss.backward()
    assert torch.isfinite(loss)
    model.eval()
    assert model.forward_inference(b).shape == (4, V)


def test_compiled_sasrec_parity(schema, batch):
    """Compiled-vs-eager logits parity (the reference OpenVINO test pattern)."""
    from replay_amd.models.nn import SasRec
    from replay_amd.models.nn.sequential.compiled import SasRecCompiled

    model = SasRec(schema, max_seq_len=L, hidden_size=16, block_count=1, dropout_rate=0.0)
    inner = model._model.eval()
    compiled = SasRecCompiled(inner, mode="batch", batch_size=4, max_seq_len=L)
    eager = inner.forward_inference(batch)
    fast = compiled.predict(batch)
    torch.testing.assert_close(eager, fast, atol=1e-5, rtol=1e-5)


def test_compiled_one_query_mode(schema, batch):
    from replay_amd.models.nn import SasRec
    from replay_amd.models.nn.sequential.compiled import SasRecCompiled

    model = SasRec(schema, max_seq_len=L, hidden_size=16, block_count=1, dropout_rate=0.0)
    compiled = SasRecCompiled(model._model.eval(), mode="one_query", max_seq_len=L)
    one = {k: v[:1] for k, v in batch.items()}
    out = compiled.predict(one)
    assert out.shape == (1, V)
    with pytest.raises(ValueError):
        compiled.predict(batch)
