"""K6 sparse-embedding path: grads restricted to touched rows, equal in
value to the dense path; cross-rank sync via explicit COO all-gather."""

import os

import pytest
import torch

pytestmark = pytest.mark.core


def test_sparse_matches_dense_gradients():
    from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
    from replay_amd.data.schema import FeatureHint, FeatureType
    from replay_amd.nn.loss import CESampled
    from replay_amd.nn.sequential.sasrec import SasRec

    V, L, E = 500, 8, 16
    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=V, embedding_dim=E,
            )
        ]
    )
    models = []
    for sparse in (False, True):
        torch.manual_seed(0)
        m = SasRec.from_params(
            schema, max_sequence_length=L, embedding_dim=E, num_blocks=1,
            num_heads=2, dropout=0.0,
            loss=CESampled(log_correction=True, vocab_size=V),
            sparse_embedding=sparse,
        )
        models.append(m)
    dense_m, sparse_m = models
    sparse_m.load_state_dict(dense_m.state_dict())

    torch.manual_seed(1)
    items = torch.randint(0, V, (4, L + 1))
    batch = {
        "item_id": items[:, :-1],
        "labels": items[:, 1:],
        "padding_mask": torch.ones(4, L, dtype=torch.bool),
        "negatives": torch.randint(0, V, (64,)),
    }
    batch["labels_padding_mask"] = batch["padding_mask"]

    l_d = dense_m(batch)
    l_s = sparse_m(batch)
    torch.testing.assert_close(l_s, l_d, rtol=1e-5, atol=1e-6)
    l_d.backward()
    l_s.backward()
    w_d = dense_m.body.embedder.embedders["item_id"].item_emb.weight
    w_s = sparse_m.body.embedder.embedders["item_id"].item_emb.weight
    assert w_s.grad.is_sparse
    torch.testing.assert_close(w_s.grad.to_dense(), w_d.grad, rtol=1e-5, atol=1e-6)


def _rank_worker(rank, world, port):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from replay_amd.parallel import sync_sparse_grads

        p = torch.nn.Parameter(torch.zeros(20, 4))
        # different nnz per rank
        idx = torch.tensor([[rank, rank + 1, 5][: rank + 2]])
        val = torch.ones(idx.shape[1], 4) * (rank + 1)
        p.grad = torch.sparse_coo_tensor(idx, val, p.shape)
        sync_sparse_grads([p])
        dense = p.grad.to_dense()
        # expected: average over ranks of each rank's scatter
        exp = torch.zeros(20, 4)
        for r in range(world):
            ridx = [r, r + 1, 5][: r + 2]
            for i in ridx:
                exp[i] += (r + 1)
        exp /= world
        torch.testing.assert_close(dense, exp)
    finally:
        dist.destroy_process_group()


def test_sync_sparse_grads_two_ranks():
    import torch.multiprocessing as mp

    port = 29771
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_rank_worker, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(120)
    assert all(p.exitcode == 0 for p in procs)


def test_trainer_with_sparse_embedding():
    """The framework Trainer + OptimizerFactory run a sparse-embedding model
    end-to-end (HybridSparseOptimizer: Adam + SparseAdam)."""
    from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
    from replay_amd.data.schema import FeatureHint, FeatureType
    from replay_amd.nn.lightning import LightningModule, OptimizerFactory
    from replay_amd.nn.lightning.optimizer import HybridSparseOptimizer
    from replay_amd.nn.loss import CESampled
    from replay_amd.nn.sequential.sasrec import SasRec
    from replay_amd.train.trainer import Trainer

    V, L, E = 200, 6, 8
    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=V, embedding_dim=E,
            )
        ]
    )
    torch.manual_seed(0)
    model = SasRec.from_params(
        schema, max_sequence_length=L, embedding_dim=E, num_blocks=1, num_heads=2,
        dropout=0.0, loss=CESampled(log_correction=True, vocab_size=V),
        sparse_embedding=True,
    )
    module = LightningModule(model, OptimizerFactory(lr=1e-2))
    opt, _ = module.configure_optimizers()
    assert isinstance(opt, HybridSparseOptimizer)

    def batches():
        torch.manual_seed(1)
        for _ in range(4):
            items = torch.randint(0, V, (4, L + 1))
            b = {
                "item_id": items[:, :-1],
                "labels": items[:, 1:],
                "padding_mask": torch.ones(4, L, dtype=torch.bool),
                "negatives": torch.randint(0, V, (32,)),
            }
            b["labels_padding_mask"] = b["padding_mask"]
            yield b

    class Loader:
        def __iter__(self):
            return batches()

        def __len__(self):
            return 4

    trainer = Trainer(max_epochs=2, accelerator="cpu", precision="32")
    trainer.fit(module, Loader())
    assert "train_loss" in trainer.logged_metrics
