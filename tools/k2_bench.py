"""K2 micro-bench: BASELINE config 5 — BinaryROC / PR-curve on 1M scores.

Compares, on one MI355X:
  - exact path (thresholds=None): HIP rocPRIM sort+scan vs torch argsort/cumsum
  - bucketized path (thresholds=1000): HIP histogram kernel
Writes one JSON line per timing.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import metrics_amd as ma
from metrics_amd import ops


def timeit(fn, warmup=3, iters=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def torch_clf_curve(preds, target):
    desc = torch.argsort(preds, descending=True)
    preds_s, target_s = preds[desc], target[desc]
    distinct = torch.where(preds_s[1:] - preds_s[:-1])[0]
    thr_idx = torch.nn.functional.pad(distinct, [0, 1], value=target_s.size(0) - 1)
    t = (target_s == 1).long()
    tps = torch.cumsum(t * 1.0, dim=0)[thr_idx]
    fps = 1 + thr_idx - tps
    return fps, tps, preds_s[thr_idx]


def main():
    torch.manual_seed(0)
    n = 1_000_000
    preds = torch.rand(n, device="cuda")
    target = torch.randint(0, 2, (n,), device="cuda")

    out = {}
    out["hip_exact_ms"] = timeit(lambda: ops.hip_binary_clf_curve(preds, target)) * 1e3
    out["torch_exact_ms"] = timeit(lambda: torch_clf_curve(preds, target)) * 1e3

    m_exact = ma.BinaryROC(thresholds=None).to("cuda")

    def roc_exact():
        m_exact.reset()
        m_exact.update(preds, target)
        m_exact.compute()

    out["roc_exact_e2e_ms"] = timeit(roc_exact, warmup=2, iters=5) * 1e3

    m_buck = ma.BinaryROC(thresholds=1000).to("cuda")

    def roc_buck():
        m_buck.reset()
        m_buck.update(preds, target)
        m_buck.compute()

    out["roc_bucketized_t1000_e2e_ms"] = timeit(roc_buck, warmup=2, iters=5) * 1e3

    # parity of the two exact formulations on this data
    f1, t1, th1 = ops.hip_binary_clf_curve(preds, target)
    f2, t2, th2 = torch_clf_curve(preds, target)
    out["exact_parity"] = bool(
        torch.equal(th1, th2) and torch.equal(t1, t2.float()) and torch.equal(f1, f2.float())
    )
    out["n"] = n
    print(json.dumps(out))


if __name__ == "__main__":
    main()
