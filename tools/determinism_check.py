"""Bit-determinism proof: identical seeded runs produce byte-identical
compute() results for the full 16-metric collection (integer atomics +
fixed-order fp64 reductions)."""
import hashlib
import sys

sys.path.insert(0, ".")
import torch
from bench import build_collection


def run():
    torch.manual_seed(777)
    coll = build_collection(1000, torch.device("cuda"), 200)
    p = [torch.randn(4096, 1000, device="cuda", dtype=torch.bfloat16) for _ in range(3)]
    t = [torch.randint(0, 1000, (4096,), device="cuda") for _ in range(3)]
    for i in range(9):
        coll.update(p[i % 3], t[i % 3])
    out = coll.compute()
    h = hashlib.sha256()
    for k in sorted(out):
        h.update(k.encode())
        h.update(out[k].cpu().contiguous().float().numpy().tobytes())
    return h.hexdigest()


if __name__ == "__main__":
    a = run()
    b = run()
    print("run1", a)
    print("run2", b)
    print("DETERMINISTIC" if a == b else "MISMATCH")
