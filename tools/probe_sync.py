import os, sys
sys.path.insert(0, ".")
import torch
import torch.distributed as dist
import metrics_amd as ma

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29772")
dist.init_process_group("nccl", rank=0, world_size=1)
m = ma.MulticlassAUROC(num_classes=7, thresholds=30).to("cuda")
p = torch.randn(64, 7, device="cuda").softmax(-1)
t = torch.randint(0, 7, (64,), device="cuda")
m.update(p, t)
print("reductions:", {k: type(v).__name__ if not isinstance(v, str) else v for k, v in m._reductions.items()})
for k in m._reductions:
    v = getattr(m, k)
    print(" state", k, type(v), getattr(v, "device", None), getattr(v, "dtype", None), getattr(v, "shape", None))
try:
    m.sync()
    print("sync ok")
except Exception as e:
    print("sync failed:", e)
dist.destroy_process_group()
