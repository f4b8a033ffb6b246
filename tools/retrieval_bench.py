"""Retrieval compute benchmark: batched segmented path vs per-query loop.

Run on a GPU box: python tools/retrieval_bench.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import metrics_amd as ma  # noqa: E402


def bench(dev: str, n: int, q: int) -> None:
    torch.manual_seed(0)
    idx = torch.randint(0, q, (n,), device=dev)
    preds = torch.rand(n, device=dev)
    target = torch.randint(0, 2, (n,), device=dev)

    def run(force_loop: bool) -> float:
        m = ma.retrieval.RetrievalMAP()
        if force_loop:
            m._batched_scores = lambda g: None
        m.update(preds, target, indexes=idx)
        if dev == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        val = m.compute()
        if dev == "cuda":
            torch.cuda.synchronize()
        return time.perf_counter() - t0, val

    t_fast, v_fast = run(False)
    t_slow, v_slow = run(True)
    assert torch.allclose(v_fast, v_slow, atol=1e-5), (v_fast, v_slow)
    print(f"{dev} n={n} queries={q}: batched {t_fast*1e3:8.1f} ms, loop {t_slow*1e3:9.1f} ms ({t_slow/t_fast:6.1f}x)")


if __name__ == "__main__":
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    for n, q in ((50_000, 5_000), (200_000, 20_000), (1_000_000, 100_000)):
        bench(dev, n, q)
