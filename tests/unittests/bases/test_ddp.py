"""Distributed state-sync tests over a 2-process gloo group (CPU cluster emulation).

Mirrors the reference's tests/unittests/bases/test_ddp.py coverage: sum/cat
sync, uneven-shape gathers, mean/max/min all-reduce fast paths, unsync
restore, custom dist_sync_fn, dist_sync_on_step.
"""
import pytest
import torch
from torch import tensor

from metrics_amd import Metric
from metrics_amd.utilities.distributed import gather_all_tensors
from tests.unittests._helpers import run_distributed


class S(Metric):
    full_state_update = False

    def __init__(self, reduce_fx="sum", default=None, **kwargs):
        super().__init__(**kwargs)
        self.add_state("x", default if default is not None else tensor(0.0), dist_reduce_fx=reduce_fx)

    def update(self, x):
        if isinstance(self.x, list):
            self.x.append(x)
        elif self._reductions["x"] == "max":
            self.x = torch.max(self.x, x)
        elif self._reductions["x"] == "min":
            self.x = torch.min(self.x, x)
        else:
            self.x = self.x + x

    def compute(self):
        from metrics_amd.utilities.data import dim_zero_cat

        if isinstance(self.x, list):
            return dim_zero_cat(self.x)
        return self.x


def _test_sum_sync(rank, world_size):
    m = S("sum")
    m.update(tensor(float(rank + 1)))
    assert m.compute() == sum(r + 1 for r in range(world_size))
    # unsync restored local state: next update continues locally
    m.update(tensor(1.0))
    m._computed = None
    assert m.compute() == sum(r + 1 for r in range(world_size)) + world_size


def _test_mean_sync(rank, world_size):
    m = S("mean")
    m.update(tensor(float(rank)))
    assert m.compute() == sum(range(world_size)) / world_size


def _test_max_min_sync(rank, world_size):
    mx = S("max", default=tensor(-float("inf")))
    mx.update(tensor(float(rank)))
    assert mx.compute() == world_size - 1
    mn = S("min", default=tensor(float("inf")))
    mn.update(tensor(float(rank)))
    assert mn.compute() == 0.0


def _test_cat_sync(rank, world_size):
    m = S("cat", default=[])
    m.update(torch.arange(2) + 10 * rank)
    out = m.compute()
    expected = torch.cat([torch.arange(2) + 10 * r for r in range(world_size)])
    assert torch.equal(out, expected), (out, expected)


def _test_cat_uneven_sync(rank, world_size):
    m = S("cat", default=[])
    m.update(torch.ones(rank + 1))
    out = m.compute()
    assert out.numel() == sum(r + 1 for r in range(world_size))


def _test_cat_some_empty(rank, world_size):
    m = S("cat", default=[])
    if rank == 0:
        m.update(torch.ones(3))
    out = m.compute()
    assert out.numel() == 3


def _test_gather_all_tensors(rank, world_size):
    t = torch.ones(rank + 1) * rank
    result = gather_all_tensors(t)
    assert len(result) == world_size
    for r in range(world_size):
        assert result[r].numel() == r + 1
        assert (result[r] == r).all()


def _test_gather_autograd(rank, world_size):
    t = torch.ones(3, requires_grad=True)
    result = gather_all_tensors(t * 2)
    # the local slot stays connected to the autograd graph
    assert result[rank].requires_grad
    result[rank].sum().backward()
    assert torch.equal(t.grad, 2 * torch.ones(3))


def _test_custom_dist_sync_fn(rank, world_size):
    calls = []

    def my_gather(t, group=None):
        calls.append(t.shape)
        return gather_all_tensors(t, group)

    m = S("sum", dist_sync_fn=my_gather)
    m.update(tensor(1.0))
    assert m.compute() == world_size
    assert len(calls) >= 1


def _test_dist_sync_on_step(rank, world_size):
    m = S("sum", dist_sync_on_step=True)
    batch_val = m(tensor(1.0))
    # forward with dist_sync_on_step returns the SYNCED batch value
    assert batch_val == world_size


def _test_custom_reduce_callable(rank, world_size):
    m = S(lambda x: x.sum(0) * 2, default=tensor(0.0))
    m.update(tensor(1.0))
    assert m.compute() == 2 * world_size


def _test_fused_bucket_many_states(rank, world_size):
    class Multi(Metric):
        full_state_update = False

        def __init__(self):
            super().__init__()
            self.add_state("a", tensor(0.0), "sum")
            self.add_state("b", torch.zeros(3, dtype=torch.long), "sum")
            self.add_state("c", tensor(0.0), "max")
            self.add_state("d", [], "cat")

        def update(self, v):
            self.a += v
            self.b += torch.ones(3, dtype=torch.long)
            self.c = torch.max(self.c, tensor(float(v)))
            self.d.append(torch.full((2,), float(v)))

        def compute(self):
            from metrics_amd.utilities.data import dim_zero_cat

            return self.a, self.b.clone(), self.c, dim_zero_cat(self.d)

    m = Multi()
    m.update(float(rank + 1))
    a, b, c, d = m.compute()
    assert a == sum(r + 1 for r in range(world_size))
    assert (b == world_size).all()
    assert c == world_size
    assert d.numel() == 2 * world_size


@pytest.mark.parametrize(
    "fn",
    [
        _test_sum_sync,
        _test_mean_sync,
        _test_max_min_sync,
        _test_cat_sync,
        _test_cat_uneven_sync,
        _test_cat_some_empty,
        _test_gather_all_tensors,
        _test_gather_autograd,
        _test_custom_dist_sync_fn,
        _test_dist_sync_on_step,
        _test_custom_reduce_callable,
        _test_fused_bucket_many_states,
    ],
)
def test_ddp(fn):
    run_distributed(fn, world_size=2)


def _test_metric_collection_ddp(rank, world_size):
    import metrics_amd as ma

    torch.manual_seed(42)
    preds = torch.randn(4, 32, 5)
    target = torch.randint(0, 5, (4, 32))
    coll = ma.MetricCollection([
        ma.MulticlassAccuracy(num_classes=5, average="micro"),
        ma.MulticlassPrecision(num_classes=5, average="macro"),
        ma.MulticlassConfusionMatrix(num_classes=5),
    ])
    # interleaved batches per rank
    for i in range(rank, 4, world_size):
        coll.update(preds[i], target[i])
    res = coll.compute()
    # compare against single-process run on ALL data
    ref_coll = ma.MetricCollection([
        ma.MulticlassAccuracy(num_classes=5, average="micro"),
        ma.MulticlassPrecision(num_classes=5, average="macro"),
        ma.MulticlassConfusionMatrix(num_classes=5),
    ])
    for i in range(4):
        ref_coll.update(preds[i], target[i])
    # disable sync for the reference (it holds all data locally)
    for m in ref_coll.values(copy_state=False):
        m.sync_on_compute = False
        m._to_sync = False
    ref = ref_coll.compute()
    for k in res:
        assert torch.allclose(res[k].float(), ref[k].float(), atol=1e-6), (k, res[k], ref[k])


def test_metric_collection_ddp():
    run_distributed(_test_metric_collection_ddp, world_size=2)


def _test_curve_metrics_ddp(rank, world_size):
    """Thresholded curve metrics (confmat sum states) must equal a single-process run."""
    import metrics_amd as ma

    torch.manual_seed(7)
    preds = torch.rand(4, 64)
    target = torch.randint(0, 2, (4, 64))
    makes = [
        lambda: ma.BinaryPrecisionRecallCurve(thresholds=25),
        lambda: ma.BinaryAUROC(thresholds=25),
        lambda: ma.BinaryAveragePrecision(thresholds=25),
        lambda: ma.BinaryROC(thresholds=25),
    ]
    for make in makes:
        m = make()
        for i in range(rank, 4, world_size):
            m.update(preds[i], target[i])
        res = m.compute()
        ref = make()
        ref.sync_on_compute = False
        for i in range(4):
            ref.update(preds[i], target[i])
        expected = ref.compute()
        if isinstance(res, tuple):
            for a, b in zip(res, expected):
                assert torch.allclose(a, b, atol=1e-6)
        else:
            assert torch.allclose(res, expected, atol=1e-6)


def _test_mc_curve_ddp(rank, world_size):
    import metrics_amd as ma

    torch.manual_seed(8)
    preds = torch.randn(4, 32, 7).softmax(-1)
    target = torch.randint(0, 7, (4, 32))
    m = ma.MulticlassAUROC(num_classes=7, thresholds=50)
    for i in range(rank, 4, world_size):
        m.update(preds[i], target[i])
    res = m.compute()
    ref = ma.MulticlassAUROC(num_classes=7, thresholds=50)
    ref.sync_on_compute = False
    for i in range(4):
        ref.update(preds[i], target[i])
    assert torch.allclose(res, ref.compute(), atol=1e-6)


def _test_pearson_welford_merge_ddp(rank, world_size):
    """Welford moment states merge across ranks via the custom parallel-merge path."""
    import metrics_amd as ma

    torch.manual_seed(9)
    x = torch.randn(4, 50)
    y = 0.5 * x + 0.3 * torch.randn(4, 50)
    m = ma.PearsonCorrCoef()
    for i in range(rank, 4, world_size):
        m.update(x[i], y[i])
    res = m.compute()
    ref = ma.PearsonCorrCoef()
    ref.sync_on_compute = False
    for i in range(4):
        ref.update(x[i], y[i])
    assert torch.allclose(res, ref.compute(), atol=1e-5)


def _test_mean_ap_ddp(rank, world_size):
    """List states (cat) in MeanAveragePrecision gather correctly."""
    import metrics_amd as ma

    torch.manual_seed(10)
    def boxes(n):
        xy = torch.rand(n, 2) * 50
        wh = torch.rand(n, 2) * 20 + 2
        return torch.cat([xy, xy + wh], 1)

    all_preds, all_tgts = [], []
    for i in range(4):
        n = 3 + i
        all_preds.append({"boxes": boxes(n), "scores": torch.rand(n), "labels": torch.randint(0, 2, (n,))})
        all_tgts.append({"boxes": boxes(3), "labels": torch.randint(0, 2, (3,))})

    m = ma.detection.MeanAveragePrecision()
    for i in range(rank, 4, world_size):
        m.update([all_preds[i]], [all_tgts[i]])
    res = m.compute()
    ref = ma.detection.MeanAveragePrecision(sync_on_compute=False)
    for i in range(4):
        ref.update([all_preds[i]], [all_tgts[i]])
    expected = ref.compute()
    assert torch.allclose(res["map"], expected["map"], atol=1e-6), (res["map"], expected["map"])
    assert torch.allclose(res["mar_100"], expected["mar_100"], atol=1e-6)


def _test_retrieval_batched_ddp(rank, world_size):
    """Retrieval list states gather; the batched compute path sees all ranks' queries."""
    import metrics_amd as ma

    torch.manual_seed(12)
    idx = torch.randint(0, 40, (400,))
    preds = torch.rand(400)
    target = torch.randint(0, 2, (400,))
    shard = slice(rank * 200, (rank + 1) * 200)
    for cls, kw in (
        (ma.retrieval.RetrievalMAP, {}),
        (ma.retrieval.RetrievalMRR, {"top_k": 3}),
        (ma.retrieval.RetrievalNormalizedDCG, {}),
    ):
        m = cls(**kw)
        m.update(preds[shard], target[shard], indexes=idx[shard])
        res = m.compute()
        # sync_on_compute must go through the constructor: _to_sync snapshots it
        ref = cls(**kw, sync_on_compute=False)
        ref.update(preds, target, indexes=idx)
        assert torch.allclose(res, ref.compute(), atol=1e-6), cls.__name__


@pytest.mark.parametrize(
    "fn",
    [_test_curve_metrics_ddp, _test_mc_curve_ddp, _test_pearson_welford_merge_ddp, _test_mean_ap_ddp,
     _test_retrieval_batched_ddp],
)
def test_ddp_metric_parity(fn):
    run_distributed(fn, world_size=2)


def _test_bench_collection_world8(rank, world_size):
    """The exact bench.py metric mix at world_size 8 (tiny shapes): the
    driver's 8-GPU scale run must be correct by construction."""
    import bench as bench_mod

    torch.manual_seed(100 + rank)
    coll = bench_mod.build_collection(13, torch.device("cpu"), curve_thresholds=10)
    preds = torch.randn(3, 16, 13)
    target = torch.randint(0, 13, (3, 16))
    for i in range(3):
        coll.update(preds[i], target[i])
    res = coll.compute()  # sync_on_compute over the world-8 group
    assert set(res) == {
        "acc_micro", "acc_macro", "precision", "recall", "f1", "fbeta2", "specificity",
        "npv", "hamming", "jaccard", "exact_match", "cohen_kappa", "mcc", "confmat",
        "auroc", "avg_precision",
    }
    for k, v in res.items():
        assert torch.isfinite(v.float()).all(), k
    # a second round: unsync restored local states, accumulation continues
    coll.update(preds[0], target[0])
    for m in coll.values(copy_state=False):
        m._computed = None
    res2 = coll.compute()
    assert torch.isfinite(res2["acc_micro"].float()).all()


def test_bench_collection_world8():
    run_distributed(_test_bench_collection_world8, world_size=8)


def _test_pooled_map_states_ddp(rank, world_size):
    # RASE keeps lazily-shaped pooled maps (scalar default -> (C,H,W) on first
    # update); PQ keeps per-category count vectors. Both must DDP-reduce to the
    # single-process result when every rank has updated at least once.
    import metrics_amd as ma

    torch.manual_seed(7)
    preds = torch.rand(4, 3, 16, 16)
    target = torch.rand(4, 3, 16, 16)
    m = ma.image.RelativeAverageSpectralError()
    for i in range(rank, 4, world_size):
        m.update(preds[i : i + 1], target[i : i + 1])
    out = m.compute()
    ref = ma.image.RelativeAverageSpectralError()
    ref.sync_on_compute = False
    ref._to_sync = False
    for i in range(4):
        ref.update(preds[i : i + 1], target[i : i + 1])
    assert torch.allclose(out, ref.compute(), atol=1e-5), (out, ref.compute())

    cats = torch.randint(0, 3, (4, 12, 12))
    inst = torch.randint(0, 2, (4, 12, 12))
    pan_t = torch.stack([cats, inst], dim=-1)
    pan_p = pan_t.clone()
    pan_p[..., 0] = (pan_p[..., 0] + (torch.rand(4, 12, 12) < 0.2).long()) % 3
    pq = ma.detection.PanopticQuality(things={0, 1}, stuffs={2})
    for i in range(rank, 4, world_size):
        pq.update(pan_p[i : i + 1], pan_t[i : i + 1])
    out = pq.compute()
    pq_ref = ma.detection.PanopticQuality(things={0, 1}, stuffs={2})
    pq_ref.sync_on_compute = False
    pq_ref._to_sync = False
    for i in range(4):
        pq_ref.update(pan_p[i : i + 1], pan_t[i : i + 1])
    assert torch.allclose(out, pq_ref.compute(), atol=1e-6), (out, pq_ref.compute())


def test_pooled_map_states_ddp():
    run_distributed(_test_pooled_map_states_ddp, world_size=2)
