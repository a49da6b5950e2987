#!/usr/bin/env python3
"""Flagship benchmark: SASRec training throughput (interactions/sec).

BASELINE config: SASRec (2 blocks, d=64, seq_len=50), bf16, ML-20M-shape
synthetic data (27278 items), random-init weights.  Weak scaling: per-GPU
batch is fixed; value is the WHOLE-JOB aggregate interactions/sec over all
ranks.

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

if "--no-tunableop" not in sys.argv:
    # rocBLAS/hipBLASLt algorithm selection from the repo-tracked tuned CSV
    # (split-K for the skinny wgrad GEMM shapes; +10% step time measured).
    # READ-ONLY by default: tuning probes on unusual shapes (the twotower
    # strided-batched loss GEMMs, bert4rec wgrad) memory-fault the GPU, so
    # probing is opt-in via --tune (run it on a throwaway box, then merge the
    # per-ordinal CSV back into tunableop_gfx950.csv).  TunableOp reads
    # per-device-ordinal files (<base><ordinal>.csv), so seed those from the
    # canonical.  Must be set before torch import.
    _tune_file = os.path.join(os.path.dirname(os.path.abspath(__file__)), "tunableop_gfx950.csv")
    if os.path.exists(_tune_file):
        import shutil as _shutil

        for _ordinal in range(8):
            _per_dev = _tune_file[: -len(".csv")] + f"{_ordinal}.csv"
            if not os.path.exists(_per_dev):
                _shutil.copy(_tune_file, _per_dev)
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1" if "--tune" in sys.argv else "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _tune_file)
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "30")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "3")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_WARMUP_DURATION_MS", "10")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_WARMUP_ITERATIONS", "1")

import torch

N_ITEMS = 27278  # ML-20M catalog size
SEQ_LEN = 50
EMB_DIM = 64
N_BLOCKS = 2
N_HEADS = 2


def apply_model_config(args):
    """--model sasrec (default, BASELINE config 2) or bert4rec (BASELINE
    config 3 shape: 4 blocks, d=128, seq_len=200; run under torchrun x8 for
    the DP=8 configuration)."""
    global SEQ_LEN, EMB_DIM, N_BLOCKS, N_HEADS
    if args.model == "bert4rec":
        SEQ_LEN, EMB_DIM, N_BLOCKS, N_HEADS = 200, 128, 4, 4
        if args.batch == 8192:  # keep tokens/step comparable (B*L = 409600)
            args.batch = 2048


def build_model(device, model_name="sasrec", loss_name="ce", sparse_embedding=False):
    from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
    from replay_amd.data.schema import FeatureHint, FeatureType
    from replay_amd.nn.loss import CE, CESampled
    from replay_amd.nn.sequential.bert4rec import Bert4Rec
    from replay_amd.nn.sequential.sasrec import SasRec

    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id",
                FeatureType.CATEGORICAL,
                is_seq=True,
                feature_hint=FeatureHint.ITEM_ID,
                cardinality=N_ITEMS,
                embedding_dim=EMB_DIM,
            )
        ]
    )
    cls = Bert4Rec if model_name == "bert4rec" else SasRec
    loss = (
        CESampled(log_correction=True, vocab_size=N_ITEMS)
        if loss_name == "sampledce"
        else CE()
    )
    kwargs = {}
    if sparse_embedding and cls is SasRec:
        kwargs["sparse_embedding"] = True
    model = cls.from_params(
        schema,
        max_sequence_length=SEQ_LEN,
        embedding_dim=EMB_DIM,
        num_blocks=N_BLOCKS,
        num_heads=N_HEADS,
        dropout=0.0,
        loss=loss,
        **kwargs,
    ).to(device)
    return model


def make_batches(n_batches, batch_size, device, seed, model_name="sasrec", n_negatives=0,
                 ragged=False):
    """Synthetic ML-20M-shape sequence batches, generated on device.

    ragged=True draws per-row history lengths (left-padded, like the real
    tokenized ML-20M sequences) so the masked attention/CE paths are in the
    timed region instead of the idealized all-valid mask."""
    gen = torch.Generator(device="cpu").manual_seed(seed)
    batches = []
    for _ in range(n_batches):
        items = torch.randint(0, N_ITEMS, (batch_size, SEQ_LEN + 1), generator=gen)
        if ragged:
            lengths = torch.randint(2, SEQ_LEN + 1, (batch_size,), generator=gen)
            pos = torch.arange(SEQ_LEN).unsqueeze(0)
            pmask = pos >= (SEQ_LEN - lengths.unsqueeze(1))  # left padding
        else:
            pmask = torch.ones(batch_size, SEQ_LEN, dtype=torch.bool)
        batch = {
            "item_id": items[:, :-1].to(device),
            "labels": items[:, 1:].to(device),
            "padding_mask": pmask.to(device),
        }
        batch["labels_padding_mask"] = batch["padding_mask"]
        if n_negatives:
            batch["negatives"] = torch.randint(
                0, N_ITEMS, (n_negatives,), generator=gen
            ).to(device)
        if model_name == "bert4rec":  # masked-token objective (15% + last)
            tm = torch.rand(batch_size, SEQ_LEN, generator=gen) < 0.15
            tm[:, -1] = True
            batch["labels"] = batch["item_id"].clone()
            batch["token_mask"] = tm.to(device)
        batches.append(batch)
    return batches


def sparse_embedding_setup(model, lr):
    """Split params for hybrid optimization: SparseAdam over sparse embedding
    tables (K6: O(touched-rows) updates at 10M+ catalogs), Adam for the rest.
    Returns (optimizers, sparse_params, ddp_ignore_names)."""
    sparse_mods = [m for m in model.modules() if isinstance(m, torch.nn.Embedding) and m.sparse]
    sparse_ids = {id(m.weight) for m in sparse_mods}
    sparse_params = [m.weight for m in sparse_mods]
    dense_params = [p for p in model.parameters() if id(p) not in sparse_ids]
    ignore = [n for n, p in model.named_parameters() if id(p) in sparse_ids]
    opts = [torch.optim.Adam(dense_params, lr=lr)]
    if sparse_params:
        opts.append(torch.optim.SparseAdam(sparse_params, lr=lr))
    return opts, sparse_params, ignore


def serve_bench(args, device, rank, world) -> None:
    """Inference benchmark: recs/sec@K with filter_seen over the full catalog
    (BASELINE config 5 shape with --items/--emb-dim overrides)."""
    import json as _json

    from replay_amd.ops.topk import catalog_topk, sharded_catalog_topk

    n_items = args.items
    emb_dim = args.emb_dim
    B = args.batch
    torch.manual_seed(7 + rank)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    # random-init catalog + queries of the named shape (no checkpoints offline)
    if world > 1:  # catalog-sharded across ranks
        shard = n_items // world
        lo = rank * shard
        items_shard = torch.randn(shard, emb_dim, device=device, dtype=dtype)
    else:
        lo = 0
        items_shard = torch.randn(n_items, emb_dim, device=device, dtype=dtype)
    queries = torch.randn(B, emb_dim, device=device, dtype=dtype)
    seen = torch.randint(0, n_items, (B, 64), device=device)

    if args.fp8 and device.type == "cuda":
        from replay_amd.ops.topk import catalog_topk_fp8, quantize_fp8

        w8, sw = quantize_fp8(items_shard)
        q8, sq = quantize_fp8(queries)

        def step():
            return catalog_topk_fp8(q8, sq, w8, sw, args.k, seen)

    else:

        def step():
            if world > 1:
                return sharded_catalog_topk(queries, items_shard, args.k, lo, seen)
            return catalog_topk(queries, items_shard, args.k, seen)

    for _ in range(args.warmup):
        step()
    if world > 1:
        torch.distributed.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if device.type == "cuda":
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], device=device, dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    value = args.steps * B / elapsed  # queries served per second (each -> K recs)
    if rank == 0:
        print(
            _json.dumps(
                {
                    "metric": f"serving recs/sec@K={args.k}, full-catalog top-K with filter_seen",
                    "value": value,
                    "unit": "queries/sec",
                    "n_gpus": world if world > 1 else args.gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "strong" if world > 1 else "weak",
                    "vs_baseline": None,
                    "dtype": ("fp8-e4m3" if args.fp8 else "bf16") if device.type == "cuda" else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": f"catalog_topk_d{emb_dim}",
                        "global_batch": B,
                        "n_items": n_items,
                        "k": args.k,
                        "parallelism": f"catalog-sharded x{world}" if world > 1 else "single",
                    },
                }
            )
        )


def twotower_bench(args, device, rank, world) -> None:
    """BASELINE config 4: Two-Tower retrieval training — 10M-item catalog,
    in-batch negatives shared across GPUs via RCCL all-gather
    (replay_amd.parallel.gather_ids inside TwoTower.forward)."""
    from replay_amd.data.nn import TensorFeatureInfo, TensorSchema
    from replay_amd.data.schema import FeatureHint, FeatureType
    from replay_amd.nn.loss import LogInCE
    from replay_amd.nn.sequential.twotower import TwoTower

    n_items = args.items
    emb_dim = args.emb_dim if args.emb_dim != 256 else 128  # config-4 default d=128
    seq_len = SEQ_LEN
    B = args.batch
    use_cuda = device.type == "cuda"
    schema = TensorSchema(
        [
            TensorFeatureInfo(
                "item_id",
                FeatureType.CATEGORICAL,
                is_seq=True,
                feature_hint=FeatureHint.ITEM_ID,
                cardinality=n_items,
                embedding_dim=emb_dim,
            )
        ]
    )
    model = TwoTower.from_params(
        schema,
        max_sequence_length=seq_len,
        embedding_dim=emb_dim,
        num_blocks=2,
        num_heads=2,
        dropout=0.0,
        loss=LogInCE(),
        sparse_embedding=not args.dense_emb,
    ).to(device)
    optimizers, sparse_params, ddp_ignore = sparse_embedding_setup(model, args.lr)
    if torch.distributed.is_initialized():
        if ddp_ignore:
            torch.nn.parallel.DistributedDataParallel._set_params_and_buffers_to_ignore_for_model(
                model, ddp_ignore
            )
        model = torch.nn.parallel.DistributedDataParallel(
            model,
            device_ids=[device.index] if use_cuda else None,
            bucket_cap_mb=64,
            gradient_as_bucket_view=True,
        )
    gen = torch.Generator(device="cpu").manual_seed(1000 + rank)
    batches = []
    for _ in range(4):
        items = torch.randint(0, n_items, (B, seq_len + 1), generator=gen)
        batch = {
            "item_id": items[:, :-1].to(device),
            "labels": items[:, 1:].to(device),
            "padding_mask": torch.ones(B, seq_len, dtype=torch.bool, device=device),
        }
        batch["labels_padding_mask"] = batch["padding_mask"]
        batches.append(batch)

    autocast = torch.autocast(device_type=device.type, dtype=torch.bfloat16, enabled=use_cuda)

    def step(i: int) -> None:
        from replay_amd.parallel import sync_sparse_grads

        with autocast:
            loss = model(batches[i % len(batches)])
        for opt in optimizers:
            opt.zero_grad(set_to_none=True)
        loss.backward()
        if world > 1 and sparse_params:
            sync_sparse_grads(sparse_params)
        for opt in optimizers:
            opt.step()

    for i in range(args.warmup):
        step(i)
    if world > 1:
        torch.distributed.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu", dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    n_gpus = world if world > 1 else args.gpus
    global_batch = B * n_gpus
    value = args.steps * global_batch * seq_len / elapsed
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "training interactions/sec, Two-Tower 10M-item retrieval",
                    "value": value,
                    "unit": "interactions/sec",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if use_cuda else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": f"twotower_2blocks_d{emb_dim}",
                        "global_batch": global_batch,
                        "seq_len": seq_len,
                        "n_items": n_items,
                        "loss": "LogInCE in-batch + cross-GPU negatives",
                        "parallelism": f"dp{n_gpus}+negatives-allgather",
                    },
                }
            )
        )


def itemknn_bench(args) -> None:
    """BASELINE config 1: ItemKNN fit/predict on an ML-1M-shape synthetic log
    (pandas/CPU plumbing path; no GPU involved)."""
    import numpy as np
    import pandas as pd

    from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
    from replay_amd.models import ItemKNN

    rng = np.random.default_rng(0)
    n_users, n_items, n_inter = 6040, 3706, 1_000_000
    # moderate popularity skew so ~1M (user, item) pairs stay distinct
    pop = 1.0 / (np.arange(n_items) + 30.0)
    draw = int(n_inter * 1.7)
    df = (
        pd.DataFrame(
            {
                "query_id": rng.integers(0, n_users, draw),
                "item_id": rng.choice(n_items, draw, p=pop / pop.sum()),
                "rating": rng.integers(1, 6, draw).astype(float),
                "timestamp": rng.integers(0, 10_000_000, draw),
            }
        )
        .drop_duplicates(["query_id", "item_id"])
        .head(n_inter)
        .reset_index(drop=True)
    )
    schema = FeatureSchema(
        [
            FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        ]
    )
    ds = Dataset(feature_schema=schema, interactions=df, categorical_encoded=True)
    model = ItemKNN(num_neighbours=100)
    t0 = time.perf_counter()
    model.fit(ds)
    fit_s = time.perf_counter() - t0
    t0 = time.perf_counter()
    recs = model.predict(ds, k=10)
    predict_s = time.perf_counter() - t0
    n_queries = ds.interactions["query_id"].nunique()
    print(
        json.dumps(
            {
                "metric": "ItemKNN fit interactions/sec + predict recs/sec@10",
                "value": len(df) / fit_s,
                "unit": "interactions/sec (fit)",
                "n_gpus": 0,
                "steps": 1,
                "warmup": 0,
                "ms_per_step": fit_s * 1000,
                "higher_is_better": True,
                "scaling": "strong",
                "vs_baseline": None,
                "dtype": "fp64",
                "data": "synthetic ML-1M shape",
                "config": {
                    "model": "itemknn_k100",
                    "n_interactions": len(df),
                    "n_queries": int(n_queries),
                    "predict_recs_per_sec": n_queries * 10 / predict_s,
                    "predict_s": predict_s,
                    "n_recs": len(recs),
                },
            }
        )
    )


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument(
        "--batch", type=int, default=None,
        help="per-GPU batch size (default: 8192 train, 1024 serve)",
    )
    parser.add_argument("--lr", type=float, default=1e-3)
    parser.add_argument("--tunableop", action="store_true", help="(default on, read-only)")
    parser.add_argument("--no-tunableop", action="store_true", help="disable rocBLAS TunableOp")
    parser.add_argument("--tune", action="store_true",
                        help="enable TunableOp PROBING (crash-prone on odd shapes; throwaway runs only)")
    parser.add_argument("--mode", choices=["train", "serve", "itemknn", "twotower"], default="train")
    parser.add_argument("--model", choices=["sasrec", "bert4rec"], default="sasrec",
                        help="train mode: sasrec (config 2) or bert4rec (config 3 shape)")
    parser.add_argument("--loss", choices=["ce", "sampledce"], default="ce",
                        help="train loss: full-softmax CE or shared-pool sampled CE (K9 fused)")
    parser.add_argument("--negatives", type=int, default=8192,
                        help="sampledce: shared negative-pool size per step")
    parser.add_argument("--ragged", action="store_true",
                        help="train: realistic variable-length (left-padded) sequences")
    parser.add_argument("--dense-emb", action="store_true",
                        help="disable sparse embedding gradients (twotower / sampledce modes)")
    parser.add_argument(
        "--graphs",
        action="store_true",
        help="capture the train step in a hipGraph (torch.cuda.CUDAGraph; single-GPU)",
    )
    parser.add_argument("--items", type=int, default=10_000_000, help="serve: catalog size")
    parser.add_argument("--emb-dim", type=int, default=256, help="serve: embedding dim")
    parser.add_argument("--k", type=int, default=100, help="serve: top-K")
    parser.add_argument("--fp8", action="store_true", help="serve: fp8 (e4m3) score GEMM")
    args = parser.parse_args()

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        # self-launch: re-exec under torchrun (one rank per GPU over RCCL) so
        # `python bench.py --gpus N` works unaided
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        cmd = [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            f"--nproc-per-node={args.gpus}",
            "--standalone",
            "--local-addr",
            "127.0.0.1",
            os.path.abspath(__file__),
            *sys.argv[1:],
        ]
        os.execvpe(cmd[0], cmd, os.environ)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    # under torchrun (WORLD_SIZE set) always init the process group, even at
    # world size 1, so RCCL init + the DDP path are exercised at every N and
    # the scaling curve's N=1 point runs the same code as N=8
    if "WORLD_SIZE" in os.environ:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        torch.distributed.init_process_group(backend="nccl" if use_cuda else "gloo")
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    torch.manual_seed(1234 + rank)

    if args.mode == "itemknn":
        itemknn_bench(args)
        return
    if args.batch is None:
        # serve: B=1024 is the max-throughput serving batch for the fused
        # top-K kernel (164K q/s vs 138K at 8192 -- the 512-row M-tiles of a
        # 16-tile-wide launch re-stream the catalog against a colder L2);
        # train: 8192 measured best for the SASRec flagship.
        args.batch = 1024 if args.mode == "serve" else 8192
    if args.mode == "serve":
        serve_bench(args, device, rank, world)
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()
        return
    if args.mode == "twotower":
        twotower_bench(args, device, rank, world)
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()
        return

    apply_model_config(args)
    if args.loss == "sampledce":
        global N_ITEMS
        N_ITEMS = args.items  # sampled CE exists for huge catalogs (config 4/5 scale)
    use_sparse = args.loss == "sampledce" and not args.dense_emb
    model = build_model(device, args.model, args.loss, sparse_embedding=use_sparse)
    optimizers, sparse_params, ddp_ignore = sparse_embedding_setup(model, args.lr)
    if torch.distributed.is_initialized():
        if ddp_ignore:
            torch.nn.parallel.DistributedDataParallel._set_params_and_buffers_to_ignore_for_model(
                model, ddp_ignore
            )
        model = torch.nn.parallel.DistributedDataParallel(
            model,
            device_ids=[device.index] if use_cuda else None,
            bucket_cap_mb=64,
            gradient_as_bucket_view=True,
        )
    if not sparse_params:
        optimizers = [
            torch.optim.Adam(
                model.parameters(), lr=args.lr, capturable=args.graphs and use_cuda and world == 1
            )
        ]
    elif args.graphs:
        raise SystemExit("--graphs is incompatible with sparse embeddings (SparseAdam)")
    optimizer = optimizers[0]  # graph-capture path uses the single dense Adam
    batches = make_batches(
        4, args.batch, device, seed=1000 + rank, model_name=args.model,
        n_negatives=args.negatives if args.loss == "sampledce" else 0,
        ragged=args.ragged,
    )

    amp_dtype = torch.bfloat16
    autocast = torch.autocast(device_type=device.type, dtype=amp_dtype, enabled=use_cuda)

    def step(i: int) -> None:
        from replay_amd.parallel import sync_sparse_grads

        batch = batches[i % len(batches)]
        with autocast:
            loss = model(batch)
        for opt in optimizers:
            opt.zero_grad(set_to_none=True)
        loss.backward()
        if world > 1 and sparse_params:
            sync_sparse_grads(sparse_params)
        for opt in optimizers:
            opt.step()

    if args.graphs and use_cuda and world == 1:
        # hipGraph capture of the whole step (HIP graphs instead of a tracing
        # compiler): one replay per step, zero per-kernel launch overhead
        static = batches[0]
        for i in range(max(3, args.warmup)):
            step(i)  # warmup + allocate grads/optimizer state
        optimizer.zero_grad(set_to_none=False)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            for p in model.parameters():
                if p.grad is not None:
                    p.grad.zero_()
            with autocast:
                static_loss = model(static)
            static_loss.backward()
            optimizer.step()
        torch.cuda.synchronize()

        def step(i: int) -> None:  # noqa: F811 — graph-replay step
            src = batches[i % len(batches)]
            static["item_id"].copy_(src["item_id"], non_blocking=True)
            static["labels"].copy_(src["labels"], non_blocking=True)
            graph.replay()

    for i in range(args.warmup):
        step(i)

    if world > 1:
        torch.distributed.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0

    if world > 1:
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu", dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if world > 1 else args.gpus
    global_batch = args.batch * n_gpus
    interactions = args.steps * global_batch * SEQ_LEN
    value = interactions / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": (
                        "training interactions/sec, SASRec ML-20M-shape"
                        if args.model == "sasrec"
                        else "training interactions/sec, BERT4Rec ML-20M-shape"
                    ),
                    "value": value,
                    "unit": "interactions/sec",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if use_cuda else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": f"{args.model}_{N_BLOCKS}blocks_d{EMB_DIM}",
                        "global_batch": global_batch,
                        "seq_len": SEQ_LEN,
                        "n_items": N_ITEMS,
                        "loss": "full-softmax CE" if args.loss == "ce" else f"sampled CE (pool {args.negatives}, log-corrected)",
                        "parallelism": f"dp{n_gpus}",
                    },
                }
            )
        )
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
