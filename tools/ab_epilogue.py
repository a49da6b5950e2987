"""Time catalog_topk at the bench config (B=1024 V=10M E=256 k=100 seen)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from replay_amd.ops.topk import catalog_topk

torch.manual_seed(7)
q = torch.randn(1024, 256, device="cuda", dtype=torch.bfloat16)
w = torch.randn(10_000_000, 256, device="cuda", dtype=torch.bfloat16)
seen = torch.randint(0, 10_000_000, (1024, 64), device="cuda")
for _ in range(8):
    catalog_topk(q, w, 100, seen)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(30):
    catalog_topk(q, w, 100, seen)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / 30 * 1000
v = os.environ.get("REPLAY_AMD_STG_VARIANT", "default")
print(f"variant={v!r}: {ms:.3f} ms ({1024/ms*1000:.0f} q/s)")
