"""Real-backend distributed sync proof on a single MI355X.

RCCL rejects two ranks on one device ("Duplicate GPU detected", rccl
init.cc:1108), so multi-rank-on-one-box splits into two halves that
together cover the whole sync engine on hardware:

1. a world-size-1 RCCL group: the fused all-reduce buckets, the gather
   path and the side-stream overlap protocol execute on the *real* RCCL
   backend (stream-ordered collectives, done events, read fences) — this is
   the half that gloo cannot emulate;
2. a 2-process gloo group exchanging *CUDA tensors*: real multi-rank
   reductions of device states (gloo stages through host) — this is the
   half world-1 RCCL cannot show.

True multi-rank RCCL runs in the driver's round-end N=1..8 scaling bench
(bench.py over torch.distributed.run). Reference behavior:
torchmetrics tests/unittests/bases/test_ddp.py:35-345.
"""
import os

import pytest
import torch
from torch import tensor

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda", 0)


# --------------------------------------------------------------- scenarios
# (module-level: spawn pickling)

def _sc_sum_unsync(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("sum").to(dev)
    m.update(tensor(float(rank + 1), device=dev))
    assert m.compute().item() == sum(r + 1 for r in range(world_size))
    # unsync restored local state; accumulation continues locally
    m.update(tensor(1.0, device=dev))
    m._computed = None
    assert m.compute().item() == sum(r + 1 for r in range(world_size)) + world_size


def _sc_mean_max_min(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("mean").to(dev)
    m.update(tensor(float(rank), device=dev))
    assert m.compute().item() == sum(range(world_size)) / world_size
    mx = S("max", default=tensor(-float("inf"))).to(dev)
    mx.update(tensor(float(rank), device=dev))
    assert mx.compute().item() == world_size - 1
    mn = S("min", default=tensor(float("inf"))).to(dev)
    mn.update(tensor(float(rank), device=dev))
    assert mn.compute().item() == 0.0


def _sc_fused_buckets_and_cat(rank, world_size):
    """Mixed-state metric: fused all-reduce buckets + gathered cat list."""
    from metrics_amd import Metric
    from metrics_amd.utilities.data import dim_zero_cat

    dev = _dev()

    class Multi(Metric):
        full_state_update = False

        def __init__(self):
            super().__init__()
            self.add_state("a", tensor(0.0), "sum")
            self.add_state("b", torch.zeros(3, dtype=torch.long), "sum")
            self.add_state("c", tensor(0.0), "max")
            self.add_state("d", [], "cat")

        def update(self, v):
            self.a = self.a + v
            self.b = self.b + torch.ones(3, dtype=torch.long, device=self.b.device)
            self.c = torch.max(self.c, tensor(float(v), device=self.c.device))
            self.d.append(torch.full((2,), float(v), device=self.a.device))

        def compute(self):
            return self.a, self.b.clone(), self.c, dim_zero_cat(self.d)

    m = Multi().to(dev)
    m.update(float(rank + 1))
    a, b, c, d = m.compute()
    assert a.item() == sum(r + 1 for r in range(world_size))
    assert (b == world_size).all()
    assert c.item() == world_size
    assert d.numel() == 2 * world_size


def _sc_uneven_cat(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("cat", default=[]).to(dev)
    m.update(torch.ones(rank + 1, device=dev))
    out = m.compute()
    assert out.numel() == sum(r + 1 for r in range(world_size))
    # some-empty corner: only rank 0 holds data
    m2 = S("cat", default=[]).to(dev)
    if rank == 0:
        m2.update(torch.ones(3, device=dev))
    out2 = m2.compute()
    assert out2.numel() == 3


def _sc_dist_sync_on_step(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("sum", dist_sync_on_step=True).to(dev)
    batch_val = m(tensor(1.0, device=dev))
    assert batch_val.item() == world_size


def _sc_collection_parity(rank, world_size):
    """The bench's metric mix, distributed == single-process on all data."""
    import metrics_amd as ma

    dev = _dev()
    torch.manual_seed(42)
    preds = torch.randn(4, 64, 13, device=dev)
    target = torch.randint(0, 13, (4, 64), device=dev)

    def make():
        kw = dict(num_classes=13, validate_args=False)
        return ma.MetricCollection({
            "acc": ma.MulticlassAccuracy(average="micro", **kw),
            "f1": ma.MulticlassF1Score(average="macro", **kw),
            "confmat": ma.MulticlassConfusionMatrix(**kw),
            "auroc": ma.MulticlassAUROC(average="macro", thresholds=20, **kw),
        }).to(dev)

    coll = make()
    for i in range(rank, 4, world_size):
        coll.update(preds[i], target[i])
    res = coll.compute()

    ref = make()
    for m in ref.values(copy_state=False):
        m.sync_on_compute = False
        m._to_sync = False
    for i in range(4):
        ref.update(preds[i], target[i])
    exp = ref.compute()
    for k in res:
        assert torch.allclose(res[k].float(), exp[k].float(), atol=1e-5), (k, res[k], exp[k])


def _sc_curve_parity(rank, world_size):
    """Thresholded curve metrics (HIP update kernels) synced across ranks."""
    import metrics_amd as ma

    dev = _dev()
    torch.manual_seed(7)
    preds = torch.rand(4, 256, device=dev)
    target = torch.randint(0, 2, (4, 256), device=dev)
    m = ma.BinaryAUROC(thresholds=25).to(dev)
    for i in range(rank, 4, world_size):
        m.update(preds[i], target[i])
    res = m.compute()
    ref = ma.BinaryAUROC(thresholds=25, sync_on_compute=False).to(dev)
    for i in range(4):
        ref.update(preds[i], target[i])
    assert torch.allclose(res, ref.compute(), atol=1e-6)


def _sc_pearson_welford(rank, world_size):
    import metrics_amd as ma

    dev = _dev()
    torch.manual_seed(9)
    x = torch.randn(4, 50, device=dev)
    y = 0.5 * x + 0.3 * torch.randn(4, 50, device=dev)
    m = ma.PearsonCorrCoef().to(dev)
    for i in range(rank, 4, world_size):
        m.update(x[i], y[i])
    res = m.compute()
    ref = ma.PearsonCorrCoef(sync_on_compute=False).to(dev)
    for i in range(4):
        ref.update(x[i], y[i])
    assert torch.allclose(res, ref.compute(), atol=1e-5)


def _sc_side_stream_overlap(rank, world_size):
    """On RCCL the fused sum sync must take the side-stream path: a done
    event is produced, compute consumes it, unsync restores local state."""
    import torch.distributed as dist

    import metrics_amd.utilities.distributed as d
    from tests.unittests.bases.test_ddp import S

    if "nccl" not in str(dist.get_backend()).lower():
        return  # only meaningful on the RCCL backend
    dev = _dev()
    assert d._use_side_stream(None), "RCCL backend must enable the side-stream sync path"
    m = S("sum").to(dev)
    m.update(tensor(float(rank + 1), device=dev))
    m.sync()
    assert m._pending_sync_event is not None, "side-stream sync should leave a pending done event"
    m._wait_pending_sync()
    assert m._pending_sync_event is None
    torch.cuda.synchronize()
    assert m.x.item() == sum(r + 1 for r in range(world_size))
    m.unsync()
    assert m.x.item() == rank + 1
    # normal compute path end-to-end (waits via _wrap_compute)
    m._computed = None
    assert m.compute().item() == sum(r + 1 for r in range(world_size))


_SCENARIOS = [
    _sc_sum_unsync,
    _sc_mean_max_min,
    _sc_fused_buckets_and_cat,
    _sc_uneven_cat,
    _sc_dist_sync_on_step,
    _sc_collection_parity,
    _sc_curve_parity,
    _sc_pearson_welford,
    _sc_side_stream_overlap,
]


def _run_all(rank, world_size):
    import torch.distributed as dist

    for fn in _SCENARIOS:
        fn(rank, world_size)
        torch.cuda.synchronize()
        dist.barrier()


def test_rccl_world1():
    """Whole scenario set on a real 1-rank RCCL communicator (in-process)."""
    import torch.distributed as dist

    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29371")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        _run_all(0, 1)
    finally:
        dist.destroy_process_group()


def test_gloo_cuda_world2():
    """Multi-rank reductions of CUDA states over a 2-process gloo group."""
    from tests.unittests._helpers import run_distributed

    run_distributed(_run_all, world_size=2, backend="gloo_cuda")


def test_rccl_world1_lazy_curve_sync():
    """sync() on a lazily-accumulated curve metric must materialize the
    confmat BEFORE the collective reads it (world-1 RCCL exercises the real
    sync path end to end)."""
    import os

    import torch.distributed as dist

    import metrics_amd as ma

    if dist.is_initialized():
        dist.destroy_process_group()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        m = ma.MulticlassAUROC(num_classes=7, thresholds=30).to("cuda")
        p = torch.randn(256, 7, device="cuda").softmax(-1)
        t = torch.randint(0, 7, (256,), device="cuda")
        m.update(p, t)
        assert m.__dict__.get("_lazy_dirty") is True
        m.sync()
        assert m.__dict__.get("_lazy_dirty") is False
        assert int(m.confmat.sum().item()) > 0
        m.unsync()
        out = m.compute()
        ref = ma.MulticlassAUROC(num_classes=7, thresholds=30)
        ref.sync_on_compute = False
        ref._to_sync = False  # CPU states must not sync over the NCCL group
        ref.update(p.cpu().float(), t.cpu())
        assert torch.allclose(out.cpu(), ref.compute(), atol=1e-5)
    finally:
        dist.destroy_process_group()
