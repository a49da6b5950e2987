import numpy as np
import pandas as pd
import pytest
import torch

from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
from replay_amd.data.schema import FeatureHint, FeatureType
from replay_amd.nn.loss import CE, CESampled, LogInCE
from replay_amd.nn.sequential.bert4rec import Bert4Rec
from replay_amd.nn.sequential.twotower import FeaturesReader, ItemTower, TwoTower
from replay_amd.nn.transform import TokenMaskTransform

pytestmark = pytest.mark.torch

V = 40
L = 12


@pytest.fixture(scope="module")
def schema():
    return TensorSchema(
        [
            TensorFeatureInfo(
                "item_id",
                FeatureType.CATEGORICAL,
                is_seq=True,
                feature_hint=FeatureHint.ITEM_ID,
                cardinality=V,
                embedding_dim=16,
            )
        ]
    )


@pytest.fixture()
def batch():
    torch.manual_seed(0)
    B = 6
    b = {
        "item_id": torch.randint(0, V, (B, L)),
        "padding_mask": torch.ones(B, L, dtype=torch.bool),
    }
    b["padding_mask"][0, :6] = False
    b["labels"] = b["item_id"].clone()
    b["labels_padding_mask"] = b["padding_mask"]
    return b


class TestBert4Rec:
    def test_train_and_inference(self, schema, batch):
        model = Bert4Rec.from_params(schema, max_sequence_length=L, embedding_dim=16, num_blocks=2, num_heads=2)
        batch = TokenMaskTransform(generator_seed=0)(batch)
        loss = model(batch)
        loss.backward()
        assert torch.isfinite(loss)
        model.eval()
        logits = model.forward_inference(batch)
        assert logits.shape == (6, V)

    def test_mask_token_is_trainable(self, schema):
        model = Bert4Rec.from_params(schema, max_sequence_length=L, embedding_dim=16)
        emb = model.body.embedder.embedders["item_id"]
        assert emb.item_emb.num_embeddings == V + 2
        assert model.mask_token == V + 1
        # padding row frozen, mask row not
        assert emb.item_emb.padding_idx == V

    def test_loss_only_on_masked(self, schema, batch):
        """With no masked position outside padding the loss ignores the rest."""
        model = Bert4Rec.from_params(schema, max_sequence_length=L, embedding_dim=16, dropout=0.0)
        b = dict(batch)
        b["token_mask"] = torch.zeros(6, L, dtype=torch.bool)
        b["token_mask"][:, -1] = True
        loss = model(b)
        assert torch.isfinite(loss)

    def test_sampled_loss(self, schema, batch):
        model = Bert4Rec.from_params(schema, max_sequence_length=L, embedding_dim=16, loss=CESampled())
        b = TokenMaskTransform(generator_seed=1)(dict(batch))
        b["negatives"] = torch.randint(0, V, (8,))
        loss = model(b)
        loss.backward()
        assert torch.isfinite(loss)


class TestTwoTower:
    def test_train_and_inference(self, schema, batch):
        model = TwoTower.from_params(schema, max_sequence_length=L, embedding_dim=16)
        loss = model(batch)
        loss.backward()
        assert torch.isfinite(loss)
        model.eval()
        logits = model.forward_inference(batch)
        assert logits.shape == (6, V)

    def test_item_tower_cache(self, schema):
        tower = ItemTower(schema, embedding_dim=16)
        tower.eval()
        emb_all = tower()
        assert emb_all.shape == (V, 16)
        assert tower._cache_valid
        sub = tower(torch.tensor([0, 3]))
        torch.testing.assert_close(sub, emb_all[[0, 3]])
        tower.train()
        emb_train = tower(torch.tensor([0, 3]))
        assert emb_train.shape == (2, 16)

    def test_item_tower_with_features(self):
        item_schema = TensorSchema(
            [
                TensorFeatureInfo(
                    "item_id", FeatureType.CATEGORICAL, is_seq=True,
                    feature_hint=FeatureHint.ITEM_ID, cardinality=V, embedding_dim=16,
                ),
                TensorFeatureInfo("genre", FeatureType.CATEGORICAL, cardinality=5, embedding_dim=4),
                TensorFeatureInfo("price", FeatureType.NUMERICAL, tensor_dim=1),
            ]
        )
        features = {
            "genre": torch.randint(0, 5, (V,)),
            "price": torch.rand(V),
        }
        tower = ItemTower.from_item_features(item_schema, features, embedding_dim=16)
        out = tower.compute_embeddings(torch.arange(V))
        assert out.shape == (V, 16)
        # buffers are in the state dict with item_reference_ keys (checkpoint contract)
        sd = tower.state_dict()
        assert "item_reference_genre" in sd and "item_reference_price" in sd

    def test_item_tower_from_checkpoint(self, tmp_path, schema):
        model = TwoTower.from_params(schema, max_sequence_length=L, embedding_dim=16)
        ckpt = {"state_dict": {f"model.{k}": v for k, v in model.state_dict().items()}}
        path = tmp_path / "tt.ckpt"
        torch.save(ckpt, path)
        tower = ItemTower.from_checkpoint(str(path), schema, embedding_dim=16)
        ids = torch.arange(5)
        torch.testing.assert_close(
            tower.compute_embeddings(ids), model.body.item_tower.compute_embeddings(ids)
        )

    def test_features_reader(self):
        item_schema = TensorSchema(
            [
                TensorFeatureInfo(
                    "item_id", FeatureType.CATEGORICAL, is_seq=True,
                    feature_hint=FeatureHint.ITEM_ID, cardinality=4, embedding_dim=8,
                ),
                TensorFeatureInfo("genre", FeatureType.CATEGORICAL, cardinality=3, embedding_dim=2),
            ]
        )
        df = pd.DataFrame({"item_id": [2, 0, 1, 3], "genre": [1, 0, 2, 1]})
        out = FeaturesReader(item_schema).read(df)
        assert out["genre"].tolist() == [0, 2, 1, 1]

    def test_get_logits_with_candidates(self, schema, batch):
        model = TwoTower.from_params(schema, max_sequence_length=L, embedding_dim=16)
        model.eval()
        q = model.get_query_embeddings(batch)
        logits = model.get_logits(q, torch.tensor([1, 2, 3]))
        assert logits.shape == (6, 3)
