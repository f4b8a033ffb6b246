"""HBM3E capacity demonstration: multiclass confusion-matrix states far past
any 80-GB-class GPU — C=100,000 keeps an 80 GB (C,C) int64 state resident on
ONE MI355X (288 GB HBM3E) with kernel-updated counts and fused compute."""
import sys, time
sys.path.insert(0, ".")
import torch
import metrics_amd as ma

for C, B in [(50_000, 2_000_000), (100_000, 2_000_000)]:
    torch.cuda.empty_cache()
    m = ma.MulticlassConfusionMatrix(num_classes=C, validate_args=False).to("cuda")
    preds = torch.randint(0, C, (B,), device="cuda")
    tgt = torch.randint(0, C, (B,), device="cuda")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    m.update(preds, tgt)
    torch.cuda.synchronize()
    t_up = time.perf_counter() - t0
    t0 = time.perf_counter()
    out = m.compute()
    torch.cuda.synchronize()
    t_cmp = time.perf_counter() - t0
    alloc = torch.cuda.memory_allocated() / 2**30
    total = int(out.sum().item())
    print(f"C={C}: state {C*C*8/2**30:.1f} GiB, allocated {alloc:.1f} GiB, "
          f"update({B} labels) {t_up*1e3:.1f} ms, compute {t_cmp*1e3:.1f} ms, sum={total} (expect {B})")
    del m, preds, tgt, out
