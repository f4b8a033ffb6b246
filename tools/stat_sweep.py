import os, time
import torch
import metrics_amd as ma

torch.manual_seed(0)
preds = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
tgt = torch.randint(0, 1000, (8192,), device="cuda")
m = ma.MulticlassAccuracy(num_classes=1000, average="macro", validate_args=False).to("cuda")
for _ in range(5):
    m.update(preds, tgt)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(100):
    m.update(preds, tgt)
torch.cuda.synchronize()
print(f"div={os.environ.get('MA_STAT_DIV','16')}: {(time.perf_counter()-t0)/100*1e6:.1f} us/update")
