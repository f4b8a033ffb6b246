"""Per-metric compute() cost on GPU for the bench collection."""
import time

import torch
from bench import build_collection

coll = build_collection(1000, torch.device("cuda"))
preds = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
tgt = torch.randint(0, 1000, (8192,), device="cuda")
coll.update(preds, tgt)
coll.compute()
torch.cuda.synchronize()

t0 = time.perf_counter()
for _ in range(10):
    coll.compute()
torch.cuda.synchronize()
print(f"collection.compute: {(time.perf_counter()-t0)/10*1e3:.2f} ms")

for name, m in coll.items(keep_base=True, copy_state=False):
    m._computed = None
torch.cuda.synchronize()
for name, m in list(coll.items(keep_base=True, copy_state=False)):
    t0 = time.perf_counter()
    for _ in range(10):
        m._computed = None
        m.compute()
    torch.cuda.synchronize()
    print(f"{name:14s} {(time.perf_counter()-t0)/10*1e3:7.3f} ms")
