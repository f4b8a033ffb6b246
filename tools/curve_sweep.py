"""Sweep curve-hist variant x c_chunk on the bench shape (run on GPU box)."""
import os
import time

import torch

assert "METRICS_AMD_CURVE_VARIANT" in os.environ
import metrics_amd as ma
from metrics_amd.ops import _hip

torch.manual_seed(0)
probs = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
tgt = torch.randint(0, 1000, (8192,), device="cuda")
m = ma.MulticlassAUROC(num_classes=1000, thresholds=200, validate_args=False).to("cuda")
for _ in range(5):
    m.update(probs, tgt)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    m.update(probs, tgt)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 50 * 1e6
print(f"variant={os.environ['METRICS_AMD_CURVE_VARIANT']} cchunk={os.environ.get('METRICS_AMD_CURVE_CCHUNK','0')}: {dt:.1f} us/update")
