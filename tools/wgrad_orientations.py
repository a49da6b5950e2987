"""Which formulation of dW = dy^T @ x (skinny outputs, huge k) is fastest?"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

N = 409600
torch.manual_seed(0)
x = torch.randn(N, 64, device="cuda", dtype=torch.bfloat16)

def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for n_out in (192, 64):
    dy = torch.randn(N, n_out, device="cuda", dtype=torch.bfloat16)
    r = {}
    r["dy.t (a) x      "] = t(lambda: dy.t() @ x)
    r["(x.t (a) dy).t  "] = t(lambda: (x.t() @ dy).t())
    C = 64
    dyc = dy.view(C, N // C, n_out)
    xc = x.view(C, N // C, 64)
    r["bmm chunk + sum "] = t(lambda: torch.baddbmm(
        torch.zeros(1, n_out, 64, device="cuda", dtype=torch.float32),
        dyc.transpose(1, 2).float(), xc.float()).sum(0) if False else
        (dyc.transpose(1, 2) @ xc).sum(0))
    r["einsum nf,ne    "] = t(lambda: torch.einsum("nf,ne->fe", dy, x))
    r["fp32 mm         "] = t(lambda: dy.t().float() @ x.float())
    best = min(r.values())
    print(f"n_out={n_out}:")
    for k, v in r.items():
        print(f"  {k}: {v:8.1f} us {'<-- best' if v == best else ''}")
