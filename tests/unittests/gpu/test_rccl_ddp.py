"""Real-RCCL distributed sync proof: 2 ranks sharing the one MI355X (cuda:0).

The gloo suite (tests/unittests/bases/test_ddp.py) proves the sync engine's
semantics on a CPU cluster emulation; this module re-exercises the same
scenarios over an actual RCCL process group so the fused all-reduce buckets,
the uneven-shape gathers AND the side-stream overlap path run on the real
backend. Reference behavior: torchmetrics tests/unittests/bases/test_ddp.py:35-345
and src/torchmetrics/utilities/distributed.py:100-153.
"""
import os

import pytest
import torch
from torch import tensor

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda", 0)


# Scenario functions must be module-level (spawn pickling).

def _rccl_sum_unsync(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("sum").to(dev)
    m.update(tensor(float(rank + 1), device=dev))
    assert m.compute().item() == sum(r + 1 for r in range(world_size))
    # unsync restored local state; accumulation continues locally
    m.update(tensor(1.0, device=dev))
    m._computed = None
    assert m.compute().item() == sum(r + 1 for r in range(world_size)) + world_size


def _rccl_mean_max_min(rank, world_size):
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


def _rccl_fused_buckets_and_cat(rank, world_size):
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


def _rccl_uneven_cat(rank, world_size):
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


def _rccl_dist_sync_on_step(rank, world_size):
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    m = S("sum", dist_sync_on_step=True).to(dev)
    batch_val = m(tensor(1.0, device=dev))
    assert batch_val.item() == world_size


def _rccl_collection_parity(rank, world_size):
    """The bench's metric mix under real RCCL == single-process on all data."""
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


def _rccl_side_stream_overlap(rank, world_size):
    """The fused sum sync must actually take the side-stream path on RCCL:
    a done event is produced, compute consumes it, and the result is right."""
    import metrics_amd.utilities.distributed as d
    from tests.unittests.bases.test_ddp import S

    dev = _dev()
    assert d._use_side_stream(None), "RCCL backend must enable the side-stream sync path"
    m = S("sum").to(dev)
    m.update(tensor(float(rank + 1), device=dev))
    # drive sync() directly so we can observe the pending event before compute
    m.sync()
    assert m._pending_sync_event is not None, "side-stream sync should leave a pending done event"
    m._wait_pending_sync()
    assert m._pending_sync_event is None
    torch.cuda.synchronize()
    assert m.x.item() == sum(r + 1 for r in range(world_size))
    m.unsync()
    assert m.x.item() == rank + 1
    # and the normal compute path end-to-end
    m._computed = None
    assert m.compute().item() == sum(r + 1 for r in range(world_size))


def _rccl_curve_parity(rank, world_size):
    """Thresholded curve metrics (HIP update kernels) synced over real RCCL."""
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


def _rccl_pearson_welford(rank, world_size):
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


_SCENARIOS = [
    _rccl_sum_unsync,
    _rccl_mean_max_min,
    _rccl_fused_buckets_and_cat,
    _rccl_uneven_cat,
    _rccl_dist_sync_on_step,
    _rccl_collection_parity,
    _rccl_side_stream_overlap,
    _rccl_curve_parity,
    _rccl_pearson_welford,
]


def _run_all(rank, world_size):
    import torch.distributed as dist

    for fn in _SCENARIOS:
        fn(rank, world_size)
        torch.cuda.synchronize()
        dist.barrier()


def test_rccl_2ranks_1gpu():
    """One spawn cost for the whole scenario set (RCCL init is seconds)."""
    from tests.unittests._helpers import run_distributed

    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    run_distributed(_run_all, world_size=2, backend="nccl")
