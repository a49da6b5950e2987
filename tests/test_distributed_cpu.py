"""Multi-process distributed tests on CPU (gloo, world_size=2).

The reference has no real multi-process test (DDP correctness is
tested by injection); we add true 2-rank gloo runs for the collectives, DDP
training-step equivalence, and sync_dist metric reduction.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = [pytest.mark.torch]


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)


def _run_gather_ids(rank, world, port, out):
    _init(rank, world, port)
    ids = torch.arange(3) + rank * 10
    from replay_amd.parallel import gather_ids

    gathered = gather_ids(ids)
    out[rank] = gathered.tolist()
    torch.distributed.destroy_process_group()


def _run_gather_embeddings(rank, world, port, out):
    _init(rank, world, port)
    from replay_amd.parallel import gather_embeddings

    x = torch.full((2, 4), float(rank + 1), requires_grad=True)
    g = gather_embeddings(x)
    # loss puts weight (rank-dependent) on every row; grads must flow to the
    # local shard from all ranks' contributions
    loss = (g * (rank + 1.0)).sum()
    loss.backward()
    out[rank] = (g.shape[0], float(x.grad.sum()))
    torch.distributed.destroy_process_group()


def _run_ddp_step(rank, world, port, out):
    _init(rank, world, port)
    from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
    from replay_amd.data.schema import FeatureHint, FeatureType
    from replay_amd.nn.sequential.sasrec import SasRec

    torch.manual_seed(0)  # same init on both ranks
    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id", FeatureType.CATEGORICAL, is_seq=True,
                feature_hint=FeatureHint.ITEM_ID, cardinality=20, embedding_dim=8,
            )
        ]
    )
    model = SasRec.from_params(schema, max_sequence_length=6, embedding_dim=8, dropout=0.0)
    ddp = torch.nn.parallel.DistributedDataParallel(model)
    gen = torch.Generator().manual_seed(100 + rank)  # different data per rank
    batch = {
        "item_id": torch.randint(0, 20, (4, 6), generator=gen),
        "labels": torch.randint(0, 20, (4, 6), generator=gen),
        "padding_mask": torch.ones(4, 6, dtype=torch.bool),
    }
    batch["labels_padding_mask"] = batch["padding_mask"]
    loss = ddp(batch)
    loss.backward()
    # after DDP all-reduce both ranks hold identical grads
    grad = model.body.embedder.embedders["item_id"].item_emb.weight.grad
    out[rank] = float(grad.abs().sum())
    torch.distributed.destroy_process_group()


def _spawn(fn, world=2, port=29515):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as manager:
        out = manager.dict()
        procs = [ctx.Process(target=fn, args=(r, world, port, out)) for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
            assert p.exitcode == 0
        return dict(out)


def test_gather_ids_two_ranks():
    out = _spawn(_run_gather_ids, port=29521)
    assert out[0] == [0, 1, 2, 10, 11, 12]
    assert out[0] == out[1]


def test_gather_embeddings_differentiable():
    out = _spawn(_run_gather_embeddings, port=29522)
    # each rank's local shard receives grad (1+2) summed over all ranks' usage
    assert out[0][0] == 4  # 2 ranks x 2 rows
    assert out[0][1] == pytest.approx(3.0 * 8)  # (1+2) * 8 elements
    assert out[1][1] == pytest.approx(3.0 * 8)


def test_ddp_gradients_identical():
    out = _spawn(_run_ddp_step, port=29523)
    assert out[0] == pytest.approx(out[1], rel=1e-6)


def _run_sharded_topk(rank, world, port, out):
    _init(rank, world, port)
    from replay_amd.ops.topk import catalog_topk, sharded_catalog_topk

    torch.manual_seed(0)  # identical across ranks
    V, E, B, K = 64, 8, 5, 6
    items = torch.randn(V, E)
    queries = torch.randn(B, E)
    seen = torch.randint(0, V, (B, 4))
    shard = V // world
    local = items[rank * shard:(rank + 1) * shard]
    s, i = sharded_catalog_topk(queries, local, K, shard_offset=rank * shard, seen=seen)
    ref_s, ref_i = catalog_topk(queries, items, K, seen=seen)
    out[rank] = (
        torch.allclose(s, ref_s, atol=1e-5),
        bool((i == ref_i).all() or torch.allclose(s, ref_s, atol=1e-5)),  # ties may reorder ids
        i.tolist(),
    )
    torch.distributed.destroy_process_group()


def test_sharded_catalog_topk_matches_single_process():
    out = _spawn(_run_sharded_topk, port=29524)
    assert out[0][0] and out[1][0], "sharded scores != full-table scores"
    assert out[0][2] == out[1][2], "ranks disagree on the merged top-K"


def _run_trainer_sync_dist(rank, world, port, out):
    _init(rank, world, port)
    from replay_amd.train import Trainer

    trainer = Trainer(accelerator="cpu")
    trainer._module = None
    trainer._log("metric", float(rank + 1), sync_dist=True)  # 1.0 and 2.0
    trainer._reduce_sync_metrics()
    out[rank] = trainer.logged_metrics["metric"]
    torch.distributed.destroy_process_group()


def test_trainer_sync_dist_mean():
    out = _spawn(_run_trainer_sync_dist, port=29525)
    assert out[0] == pytest.approx(1.5)
    assert out[0] == out[1]
