"""Offline TunableOp probe for the bert4rec projection/FFN GEMM shapes.

Runs ONLY the plain forward/dgrad GEMM shapes (the round-1 crash came from
tuning probes on the padded-ld wgrad / strided-batched shapes, so those are
excluded here).  Writes tuned results to the per-ordinal CSV, which we then
merge into the canonical tunableop_gfx950.csv.
"""

import os
import sys

os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = sys.argv[1] if len(sys.argv) > 1 else "/tmp/tune_bert.csv"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS"] = "50"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS"] = "5"

import torch

N = 409600
E = 128
shapes_fwd = [
    (N, E, 3 * E),   # in_proj
    (N, E, E),       # out_proj / ffn w1 / w2 (hidden == E)
    (N, 3 * E, E),   # dgrad in_proj
]
torch.manual_seed(0)
for (m, k, n) in shapes_fwd:
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        y = torch.nn.functional.linear(a, w, b)   # addmm path (GemmAndBias)
        y2 = a @ w.t()                            # plain NT
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(10):
        y = torch.nn.functional.linear(a, w, b)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 10 * 1000
    print(f"tuned linear m={m} k={k} n={n}: {ms:.3f} ms  ({2*m*k*n/ms*1e-9:.0f} GF/s)")
print("done")
