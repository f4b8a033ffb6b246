import sys, time
sys.path.insert(0, ".")
import torch
from bench import build_collection

coll = build_collection(1000, torch.device("cuda"), 200)
p = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
t = torch.randint(0, 1000, (8192,), device="cuda")
for _ in range(3):
    coll.update(p, t)
coll.compute()
torch.cuda.synchronize()

def bust():
    for m in coll.values(copy_state=False):
        m._computed = None

n = 30
bust(); torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(n):
    bust()
    coll.compute()
t_py = (time.perf_counter() - t0) / n  # python+enqueue (async tail hidden)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(n):
    bust()
    coll.compute()
    torch.cuda.synchronize()
t_wall = (time.perf_counter() - t0) / n
print(f"compute python+enqueue: {t_py*1e6:.0f} us   wall(synced): {t_wall*1e6:.0f} us")

import cProfile, pstats, io
pr = cProfile.Profile()
pr.enable()
for _ in range(n):
    bust()
    coll.compute()
pr.disable()
st = pstats.Stats(pr); st.sort_stats("cumulative")
buf = io.StringIO(); st.stream = buf; st.print_stats(34)
print("\n".join(buf.getvalue().splitlines()[4:30]))
