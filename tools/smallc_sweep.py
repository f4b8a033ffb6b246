"""Small-C stat-kernel sweep: thread-per-row (tiny) vs wave-per-row variants.

Run on a GPU box:  MA_STAT_SMALLC=0 python tools/smallc_sweep.py   (wave only)
                   MA_STAT_SMALLC=128 python tools/smallc_sweep.py (auto tiny, default)
Prints us/update for several (B, C) shapes and checks tiny == wave numerics.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import metrics_amd as ma  # noqa: E402

torch.manual_seed(0)
SHAPES = [(65536, 8), (65536, 16), (65536, 37), (65536, 48), (32768, 64), (32768, 128), (16384, 256), (8192, 1000)]


def bench_one(B, C):
    preds = torch.randn(B, C, device="cuda", dtype=torch.bfloat16)
    tgt = torch.randint(0, C, (B,), device="cuda")
    m = ma.MulticlassAccuracy(num_classes=C, average="macro", validate_args=False).to("cuda")
    for _ in range(5):
        m.update(preds, tgt)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(100):
        m.update(preds, tgt)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 100 * 1e6
    # numerics vs plain torch fp32
    m.reset()
    m.update(preds, tgt)
    mine = m.compute().item()
    am = preds.float().argmax(1)
    per_class = torch.zeros(C, device="cuda")
    for c in range(min(C, 64)):
        sel = tgt == c
        per_class[c] = (am[sel] == c).float().mean() if sel.any() else float("nan")
    ref = per_class[: min(C, 64)].nanmean().item() if C <= 64 else None
    return us, mine, ref


if __name__ == "__main__":
    print(f"MA_STAT_SMALLC={os.environ.get('MA_STAT_SMALLC', '128')}")
    for B, C in SHAPES:
        us, mine, ref = bench_one(B, C)
        ok = "" if ref is None else f" ref={ref:.5f} {'OK' if abs(mine - ref) < 1e-4 else 'MISMATCH'}"
        print(f"B={B} C={C}: {us:7.1f} us/update  acc={mine:.5f}{ok}")
