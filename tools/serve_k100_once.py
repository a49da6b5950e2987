import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from replay_amd.ops.topk import catalog_topk
torch.manual_seed(7)
q = torch.randn(1024, 256, device="cuda", dtype=torch.bfloat16)
w = torch.randn(10_000_000, 256, device="cuda", dtype=torch.bfloat16)
seen = torch.randint(0, 10_000_000, (1024, 64), device="cuda")
for _ in range(3):
    catalog_topk(q, w, 100, seen)
torch.cuda.synchronize()
for _ in range(5):
    catalog_topk(q, w, 100, seen)
torch.cuda.synchronize()
print("done")
