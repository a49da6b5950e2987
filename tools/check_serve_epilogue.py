"""GPU A/B + parity for the batched-atomic serve epilogue (all E variants)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from replay_amd.ops.topk import catalog_topk

def parity(M, V, E, k=10, seed=0):
    torch.manual_seed(seed)
    q = torch.randn(M, E, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(V, E, device="cuda", dtype=torch.bfloat16) * 0.05
    v, i = catalog_topk(q, w, k)
    rv, ri = torch.topk(q.float() @ w.float().t(), k, dim=1)
    bad = 0
    for r in range(M):
        if not torch.equal(torch.sort(i[r])[0], torch.sort(ri[r])[0]):
            # allow tie-at-boundary rows
            vs, rs = torch.sort(v[r].float())[0], torch.sort(rv[r].float())[0]
            if not torch.allclose(vs, rs, rtol=2e-2, atol=2e-2):
                bad += 1
    print(f"parity M={M} V={V} E={E}: mismatched rows = {bad}")
    return bad

total = 0
total += parity(1024, 1_000_000, 256)
total += parity(513, 80_001, 256, k=20, seed=1)
total += parity(1024, 500_000, 128, seed=2)
total += parity(1024, 500_000, 64, seed=3)
assert total == 0, f"{total} mismatched rows"

def timeit(M, V, E, iters=20):
    torch.manual_seed(0)
    q = torch.randn(M, E, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(V, E, device="cuda", dtype=torch.bfloat16) * 0.05
    for _ in range(5):
        catalog_topk(q, w, 10)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        catalog_topk(q, w, 10)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1000
    print(f"time M={M} V={V} E={E}: {ms:.3f} ms  ({M/ms*1000:.0f} q/s)")

timeit(1024, 10_000_000, 256)
timeit(1024, 10_000_000, 128)
timeit(1024, 10_000_000, 64)
print("OK")
