"""Deep re-tune of the skinny K=409600 wgrad GEMM shapes of the d=64
flagship (dW = dy^T @ x).  The round-1 quick probe (50 ms / 5 iters)
settled on non-split-K solutions at ~340 us where the read floor is
~35 us; give the tuner a much larger budget so GSU/split-K solutions are
explored."""
import os, sys

out = sys.argv[1] if len(sys.argv) > 1 else "/tmp/tune_wgrad.csv"
os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = out
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS"] = "500"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS"] = "100"

import time
import torch

N = 409600
torch.manual_seed(0)
x = torch.randn(N, 64, device="cuda", dtype=torch.bfloat16)
for n_out in (192, 64):
    dy = torch.randn(N, n_out, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        dw = dy.t() @ x            # nt_64_<n_out>_409600
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        dw = dy.t() @ x
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 20 * 1000
    print(f"wgrad [{n_out},64] k={N}: {ms*1000:.1f} us")
# also the ffn conv1d wgrads if they differ in layout: dy [N,64] x [N,64]
# covered by n_out=64 above.  And the CE dweight (V=27278, k=409600):
h = x
dl = torch.randn(N, 27328, device="cuda", dtype=torch.bfloat16)[:, :27278]
for _ in range(2):
    dw2 = dl.t() @ h               # nt_64_27278_409600_ld_64_27328_64
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(5):
    dw2 = dl.t() @ h
torch.cuda.synchronize()
print(f"ce dweight: {(time.perf_counter()-t0)/5*1000:.3f} ms")
print("done")
