"""Per-metric GPU kernel attribution for the bench collection (run on MI355X)."""
import torch
import metrics_amd as ma
from bench import build_collection

coll = build_collection(1000, torch.device("cuda"))
preds = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
tgt = torch.randint(0, 1000, (8192,), device="cuda")

for name, m in list(coll.items(keep_base=True, copy_state=False)):
    m.update(preds, tgt)  # warm
torch.cuda.synchronize()

from torch.profiler import profile, ProfilerActivity

for name, m in list(coll.items(keep_base=True, copy_state=False)):
    with profile(activities=[ProfilerActivity.CUDA], record_shapes=False) as prof:
        for _ in range(5):
            m.update(preds, tgt)
        torch.cuda.synchronize()
    evs = [e for e in prof.key_averages() if e.device_type == torch.autograd.DeviceType.CUDA or e.self_device_time_total > 0]
    total = sum(e.self_device_time_total for e in prof.key_averages())
    tops = sorted(prof.key_averages(), key=lambda e: -e.self_device_time_total)[:4]
    desc = "; ".join(f"{e.key.split('<')[0][:40]}x{e.count}:{e.self_device_time_total/5:.0f}us" for e in tops if e.self_device_time_total > 0)
    print(f"{name:14s} {total/5:7.1f} us/update | {desc}")
