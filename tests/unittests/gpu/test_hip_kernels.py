"""GPU numerics tests: every HIP kernel vs the plain-torch fp32 CPU reference.

All tests marked ``gpu`` — run on an MI355X via gpurun / the driver.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

import metrics_amd as ma
from metrics_amd import ops


@pytest.fixture(autouse=True)
def _require_hip():
    assert torch.cuda.is_available()
    assert ops.hip_available(), "HIP kernel library must be built and loadable on a GPU box"


def test_bincount_matches_torch():
    x = torch.randint(0, 1000, (1_000_000,), device="cuda")
    out = ops.hip_bincount(x, 1000)
    ref = torch.bincount(x.cpu(), minlength=1000)
    assert torch.equal(out.cpu(), ref)


def test_bincount_large_bins():
    x = torch.randint(0, 100_000, (2_000_000,), device="cuda")
    out = ops.hip_bincount(x, 100_000)
    ref = torch.bincount(x.cpu(), minlength=100_000)
    assert torch.equal(out.cpu(), ref)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("C", [10, 1000, 1001])
def test_mc_stat_logits_kernel(dtype, C):
    B = 4096
    preds = torch.randn(B, C, device="cuda", dtype=dtype)
    target = torch.randint(0, C, (B,), device="cuda")
    tp, fp, tn, fn, confmat = ops.multiclass_stat_scores_fused(preds, target, C, None, want_confmat=True)
    # CPU fp32 reference
    plabels = preds.float().cpu().argmax(-1)
    tcpu = target.cpu()
    ref_cm = torch.zeros(C, C, dtype=torch.long)
    for t, p in zip(tcpu.tolist(), plabels.tolist()):
        ref_cm[t, p] += 1
    assert torch.equal(confmat.cpu(), ref_cm)
    assert torch.equal(tp.cpu(), ref_cm.diag())
    assert torch.equal(fp.cpu(), ref_cm.sum(0) - ref_cm.diag())
    assert torch.equal(fn.cpu(), ref_cm.sum(1) - ref_cm.diag())
    assert torch.equal(tn.cpu(), ref_cm.sum() - ref_cm.diag() - (ref_cm.sum(0) - ref_cm.diag()) - (ref_cm.sum(1) - ref_cm.diag()))


def test_mc_stat_logits_argmax_tiebreak():
    """torch.argmax returns the first max index; the kernel must match."""
    B, C = 512, 37
    preds = torch.randint(0, 3, (B, C), device="cuda").float()  # many ties
    target = torch.randint(0, C, (B,), device="cuda")
    tp, fp, tn, fn, cm = ops.multiclass_stat_scores_fused(preds, target, C, None, want_confmat=True)
    plabels = preds.cpu().argmax(-1)
    ref_cm = torch.zeros(C, C, dtype=torch.long)
    for t, p in zip(target.cpu().tolist(), plabels.tolist()):
        ref_cm[t, p] += 1
    assert torch.equal(cm.cpu(), ref_cm)


def test_mc_stat_ignore_index():
    B, C = 2048, 100
    preds = torch.randn(B, C, device="cuda")
    target = torch.randint(0, C, (B,), device="cuda")
    target[::5] = -1
    tp, fp, tn, fn, _ = ops.multiclass_stat_scores_fused(preds, target, C, -1, want_confmat=False)
    keep = target != -1
    plabels = preds[keep].cpu().argmax(-1)
    t = target[keep].cpu()
    assert tp.sum().item() == (plabels == t).sum().item()
    assert (tp + fn).sum().item() == keep.sum().item()


def test_binary_stat_kernel_probs_and_logits():
    N = 1_000_000
    target = torch.randint(0, 2, (N,), device="cuda")
    # probabilities in [0,1]: raw thresholding
    probs = torch.rand(N, device="cuda")
    tp, fp, tn, fn = ops.binary_stat_scores_fused(probs, target, 0.5, None)
    ref_p = probs > 0.5
    assert tp.item() == ((ref_p == 1) & (target == 1)).sum().item()
    assert tn.item() == ((ref_p == 0) & (target == 0)).sum().item()
    # logits outside [0,1]: sigmoid thresholding must be picked on-device
    logits = torch.randn(N, device="cuda") * 3
    tp2, fp2, tn2, fn2 = ops.binary_stat_scores_fused(logits, target, 0.5, None)
    ref_p2 = torch.sigmoid(logits) > 0.5
    assert tp2.item() == ((ref_p2 == 1) & (target == 1)).sum().item()
    assert fn2.item() == ((ref_p2 == 0) & (target == 1)).sum().item()


def test_multilabel_stat_kernel():
    N, L = 65536, 20
    preds = torch.rand(N, L, device="cuda")
    target = torch.randint(0, 2, (N, L), device="cuda")
    tp, fp, tn, fn = ops.multilabel_stat_scores_fused(preds, target, 0.5, None)
    p = preds > 0.5
    assert torch.equal(tp.cpu(), ((p == 1) & (target == 1)).sum(0).cpu())
    assert torch.equal(fp.cpu(), ((p == 1) & (target == 0)).sum(0).cpu())


def test_binary_curve_confmat_kernel():
    N, T = 1_000_000, 1000
    preds = torch.rand(N, device="cuda")
    target = torch.randint(0, 2, (N,), device="cuda")
    thr = torch.linspace(0, 1, T, device="cuda")
    cm = ops.binary_curve_confmat(preds, target, thr, None)
    # CPU reference via the bincount formulation on a subsample of thresholds
    cm_ref = ops.binary_curve_confmat(preds.cpu(), target.cpu(), thr.cpu(), None)
    assert torch.equal(cm.cpu(), cm_ref)


def test_multiclass_curve_confmat_kernel():
    B, C, T = 8192, 50, 200
    probs = torch.randn(B, C, device="cuda").softmax(-1)
    target = torch.randint(0, C, (B,), device="cuda")
    thr = torch.linspace(0, 1, T, device="cuda")
    cm = ops.multiclass_curve_confmat(probs, target, thr, None)
    cm_ref = ops.multiclass_curve_confmat(probs.cpu(), target.cpu(), thr.cpu(), None)
    assert torch.equal(cm.cpu(), cm_ref)


def test_multilabel_curve_confmat_kernel():
    B, L, T = 8192, 10, 100
    probs = torch.rand(B, L, device="cuda")
    target = torch.randint(0, 2, (B, L), device="cuda")
    thr = torch.linspace(0, 1, T, device="cuda")
    cm = ops.multilabel_curve_confmat(probs, target, thr, None)
    cm_ref = ops.multilabel_curve_confmat(probs.cpu(), target.cpu(), thr.cpu(), None)
    assert torch.equal(cm.cpu(), cm_ref)


def test_err_reduce_kernel():
    N = 1_000_000
    x = torch.randn(N, device="cuda")
    y = torch.randn(N, device="cuda")
    for op in ("sq_err", "abs_err", "sq_log_err"):
        if op == "sq_log_err":
            x_, y_ = x.abs(), y.abs()
        else:
            x_, y_ = x, y
        out = ops.err_reduce_sum(x_, y_, op)
        ref = ops.err_reduce_sum(x_.cpu(), y_.cpu(), op)
        assert torch.allclose(out.cpu(), ref, rtol=1e-10, atol=1e-6), (op, out, ref)
    mom = ops.err_reduce_sum(x, y, "moments")
    ref = ops.err_reduce_sum(x.cpu(), y.cpu(), "moments")
    assert torch.allclose(mom.cpu(), ref, rtol=1e-10, atol=1e-5)


def test_err_reduce_deterministic():
    x = torch.randn(3_000_000, device="cuda")
    y = torch.randn(3_000_000, device="cuda")
    a = ops.err_reduce_sum(x, y, "sq_err")
    b = ops.err_reduce_sum(x, y, "sq_err")
    assert torch.equal(a, b)


def test_box_iou_kernel():
    N, M = 500, 700
    def rand_boxes(n):
        xy = torch.rand(n, 2, device="cuda") * 100
        wh = torch.rand(n, 2, device="cuda") * 50 + 1
        return torch.cat([xy, xy + wh], dim=1)
    b1, b2 = rand_boxes(N), rand_boxes(M)
    for variant in ("iou", "giou", "diou", "ciou"):
        out = ops.box_iou_pairwise(b1, b2, variant)
        ref = ops.box_iou_pairwise(b1.cpu(), b2.cpu(), variant)
        assert torch.allclose(out.cpu(), ref, atol=1e-4), variant


def test_metric_end_to_end_gpu_vs_cpu():
    """Full modular metrics on GPU (HIP path) must match the CPU torch path."""
    torch.manual_seed(0)
    preds = torch.randn(4096, 100)
    target = torch.randint(0, 100, (4096,))
    for make in (
        lambda: ma.MulticlassAccuracy(num_classes=100, average="micro"),
        lambda: ma.MulticlassAccuracy(num_classes=100, average="macro"),
        lambda: ma.MulticlassF1Score(num_classes=100, average="weighted"),
        lambda: ma.MulticlassConfusionMatrix(num_classes=100),
    ):
        m_cpu, m_gpu = make(), make().to("cuda")
        m_cpu.update(preds, target)
        m_gpu.update(preds.cuda(), target.cuda())
        assert torch.allclose(m_cpu.compute().float(), m_gpu.compute().float().cpu(), atol=1e-6)

    bp, bt = torch.rand(100_000), torch.randint(0, 2, (100_000,))
    for make in (
        lambda: ma.BinaryAccuracy(),
        lambda: ma.BinaryF1Score(),
        lambda: ma.BinaryAUROC(thresholds=1000),
        lambda: ma.BinaryPrecisionRecallCurve(thresholds=500),
    ):
        m_cpu, m_gpu = make(), make().to("cuda")
        r_cpu = m_cpu(bp, bt)
        r_gpu = m_gpu(bp.cuda(), bt.cuda())
        if isinstance(r_cpu, tuple):
            for a, b in zip(r_cpu, r_gpu):
                assert torch.allclose(a, b.cpu(), atol=1e-6)
        else:
            assert torch.allclose(r_cpu, r_gpu.cpu(), atol=1e-6)


def test_bf16_inputs_gpu():
    preds = torch.randn(8192, 1000, device="cuda", dtype=torch.bfloat16)
    target = torch.randint(0, 1000, (8192,), device="cuda")
    m = ma.MulticlassAccuracy(num_classes=1000, average="micro").to("cuda")
    v = m(preds, target)
    ref = (preds.float().argmax(-1) == target).float().mean()
    assert torch.allclose(v, ref, atol=1e-6)


def test_into_state_stat_paths_match_cpu():
    """The accumulate-into-state GPU update must equal the CPU torch path."""
    torch.manual_seed(5)
    preds = torch.randn(2048, 64)
    target = torch.randint(0, 64, (2048,))
    target[::9] = -1
    for avg in ("micro", "macro", None):
        m_cpu = ma.MulticlassStatScores(num_classes=64, average=avg, ignore_index=-1)
        m_gpu = ma.MulticlassStatScores(num_classes=64, average=avg, ignore_index=-1).to("cuda")
        for chunk in range(4):
            sl = slice(chunk * 512, (chunk + 1) * 512)
            m_cpu.update(preds[sl], target[sl])
            m_gpu.update(preds[sl].cuda(), target[sl].cuda())
        assert torch.equal(m_cpu.tp, m_gpu.tp.cpu())
        assert torch.equal(m_cpu.tn, m_gpu.tn.cpu())
        assert torch.allclose(m_cpu.compute().float(), m_gpu.compute().float().cpu(), atol=1e-6)


def test_into_state_curve_paths_match_cpu():
    torch.manual_seed(6)
    # binary with ignore index
    bp = torch.rand(50_000)
    bt = torch.randint(0, 2, (50_000,))
    bt[::11] = -1
    m_cpu = ma.BinaryPrecisionRecallCurve(thresholds=100, ignore_index=-1)
    m_gpu = ma.BinaryPrecisionRecallCurve(thresholds=100, ignore_index=-1).to("cuda")
    m_cpu.update(bp, bt)
    m_gpu.update(bp.cuda(), bt.cuda())
    m_gpu._maybe_flush_lazy()  # raw state read below; public APIs flush themselves
    assert torch.equal(m_cpu.confmat, m_gpu.confmat.cpu())

    # multiclass AUROC thresholded
    probs = torch.randn(4096, 37).softmax(-1)
    tgt = torch.randint(0, 37, (4096,))
    a_cpu = ma.MulticlassAUROC(num_classes=37, thresholds=200)
    a_gpu = ma.MulticlassAUROC(num_classes=37, thresholds=200).to("cuda")
    a_cpu.update(probs, tgt)
    a_gpu.update(probs.cuda(), tgt.cuda())
    a_gpu._maybe_flush_lazy()
    assert torch.equal(a_cpu.confmat, a_gpu.confmat.cpu())
    assert torch.allclose(a_cpu.compute(), a_gpu.compute().cpu(), atol=1e-6)

    # multilabel AP thresholded
    mlp = torch.rand(4096, 12)
    mlt = torch.randint(0, 2, (4096, 12))
    p_cpu = ma.MultilabelAveragePrecision(num_labels=12, thresholds=50)
    p_gpu = ma.MultilabelAveragePrecision(num_labels=12, thresholds=50).to("cuda")
    p_cpu.update(mlp, mlt)
    p_gpu.update(mlp.cuda(), mlt.cuda())
    p_gpu._maybe_flush_lazy()
    assert torch.equal(p_cpu.confmat, p_gpu.confmat.cpu())


def test_multilabel_fused_with_logits():
    torch.manual_seed(7)
    logits = torch.randn(8192, 30) * 2
    tgt = torch.randint(0, 2, (8192, 30))
    m_cpu = ma.MultilabelF1Score(num_labels=30, average="macro")
    m_gpu = ma.MultilabelF1Score(num_labels=30, average="macro").to("cuda")
    m_cpu.update(logits, tgt)
    m_gpu.update(logits.cuda(), tgt.cuda())
    assert torch.allclose(m_cpu.compute(), m_gpu.compute().cpu(), atol=1e-6)


def test_bincount_deterministic_repeat():
    x = torch.randint(0, 777, (4_000_000,), device="cuda")
    a = ops.hip_bincount(x, 777)
    b = ops.hip_bincount(x, 777)
    assert torch.equal(a, b)


def test_curve_in_kernel_softmax_logits():
    """Raw logits to curve metrics: the kernel softmaxes in-flight; values must
    match the CPU torch path to float tolerance (sum-order differs by design),
    and be deterministic across runs."""
    torch.manual_seed(12)
    logits = torch.randn(4096, 50, device="cuda") * 3
    tgt = torch.randint(0, 50, (4096,), device="cuda")
    a1 = ma.MulticlassAUROC(num_classes=50, thresholds=100).to("cuda")
    a2 = ma.MulticlassAUROC(num_classes=50, thresholds=100).to("cuda")
    a1.update(logits, tgt)
    a2.update(logits, tgt)
    a1._maybe_flush_lazy()
    a2._maybe_flush_lazy()
    assert torch.equal(a1.confmat, a2.confmat)  # deterministic
    cpu = ma.MulticlassAUROC(num_classes=50, thresholds=100)
    cpu.update(logits.cpu(), tgt.cpu())
    # confmat counts can differ by ulp-at-threshold cases; the metric must agree
    assert torch.allclose(a1.compute().cpu(), cpu.compute(), atol=1e-3)
    diff = (a1.confmat.cpu() - cpu.confmat).abs().sum().item()
    assert diff <= 8, f"confmat count drift too large: {diff}"
    # pre-normalized probabilities keep exact equality (flag never set)
    probs = logits.softmax(-1)
    g = ma.MulticlassAUROC(num_classes=50, thresholds=100).to("cuda")
    c = ma.MulticlassAUROC(num_classes=50, thresholds=100)
    g.update(probs, tgt)
    c.update(probs.cpu(), tgt.cpu())
    g._maybe_flush_lazy()
    assert torch.equal(g.confmat.cpu(), c.confmat)


def test_curve_in_kernel_sigmoid_logits():
    torch.manual_seed(13)
    logits = torch.randn(100_000, device="cuda") * 4
    tgt = torch.randint(0, 2, (100_000,), device="cuda")
    g = ma.BinaryROC(thresholds=64).to("cuda")
    c = ma.BinaryROC(thresholds=64)
    g.update(logits, tgt)
    c.update(logits.cpu(), tgt.cpu())
    g._maybe_flush_lazy()
    diff = (g.confmat.cpu() - c.confmat).abs().sum().item()
    assert diff <= 4, diff
    # multilabel sigmoid
    ml_logits = torch.randn(8192, 12, device="cuda") * 4
    ml_tgt = torch.randint(0, 2, (8192, 12), device="cuda")
    gm = ma.MultilabelAveragePrecision(num_labels=12, thresholds=50).to("cuda")
    cm = ma.MultilabelAveragePrecision(num_labels=12, thresholds=50)
    gm.update(ml_logits, ml_tgt)
    cm.update(ml_logits.cpu(), ml_tgt.cpu())
    assert torch.allclose(gm.compute().cpu(), cm.compute(), atol=1e-3)


def test_pooled_hist_interleaved_metrics():
    """Two metrics with the same (C,T) share the pooled hist scratch; the
    suffix kernel re-zeroes it in-flight, so interleaved updates must not leak."""
    torch.manual_seed(14)
    p1 = torch.rand(2048, 20, device="cuda")
    p1 = p1 / p1.sum(-1, keepdim=True)
    t1 = torch.randint(0, 20, (2048,), device="cuda")
    p2 = torch.rand(2048, 20, device="cuda")
    p2 = p2 / p2.sum(-1, keepdim=True)
    t2 = torch.randint(0, 20, (2048,), device="cuda")
    m1 = ma.MulticlassPrecisionRecallCurve(num_classes=20, thresholds=30).to("cuda")
    m2 = ma.MulticlassPrecisionRecallCurve(num_classes=20, thresholds=30).to("cuda")
    for _ in range(3):
        m1.update(p1, t1)
        m2.update(p2, t2)
    r1 = ma.MulticlassPrecisionRecallCurve(num_classes=20, thresholds=30)
    r2 = ma.MulticlassPrecisionRecallCurve(num_classes=20, thresholds=30)
    for _ in range(3):
        r1.update(p1.cpu(), t1.cpu())
        r2.update(p2.cpu(), t2.cpu())
    m1._maybe_flush_lazy()
    m2._maybe_flush_lazy()
    assert torch.equal(m1.confmat.cpu(), r1.confmat)
    assert torch.equal(m2.confmat.cpu(), r2.confmat)


def test_confmat_into_state_repeated():
    torch.manual_seed(15)
    m_gpu = ma.MulticlassConfusionMatrix(num_classes=33).to("cuda")
    m_cpu = ma.MulticlassConfusionMatrix(num_classes=33)
    for _ in range(5):
        preds = torch.randn(1024, 33)
        tgt = torch.randint(0, 33, (1024,))
        m_gpu.update(preds.cuda(), tgt.cuda())
        m_cpu.update(preds, tgt)
    assert torch.equal(m_gpu.confmat.cpu(), m_cpu.confmat)
    m_gpu.reset()
    m_cpu.reset()
    preds = torch.randn(512, 33)
    tgt = torch.randint(0, 33, (512,))
    m_gpu.update(preds.cuda(), tgt.cuda())
    m_cpu.update(preds, tgt)
    assert torch.equal(m_gpu.confmat.cpu(), m_cpu.confmat)


def test_exact_match_fused_gpu():
    torch.manual_seed(16)
    m_gpu = ma.MulticlassExactMatch(num_classes=40).to("cuda")
    m_cpu = ma.MulticlassExactMatch(num_classes=40)
    for _ in range(4):
        preds = torch.randn(1024, 40)
        tgt = torch.randint(0, 40, (1024,))
        m_gpu.update(preds.cuda(), tgt.cuda())
        m_cpu.update(preds, tgt)
    assert torch.equal(m_gpu.correct.cpu(), m_cpu.correct)
    assert torch.equal(m_gpu.total.cpu(), m_cpu.total)
    # with ignore_index
    gi = ma.MulticlassExactMatch(num_classes=40, ignore_index=-1).to("cuda")
    ci = ma.MulticlassExactMatch(num_classes=40, ignore_index=-1)
    preds = torch.randn(2048, 40)
    tgt = torch.randint(0, 40, (2048,))
    tgt[::5] = -1
    gi.update(preds.cuda(), tgt.cuda())
    ci.update(preds, tgt)
    assert torch.equal(gi.correct.cpu(), ci.correct) and torch.equal(gi.total.cpu(), ci.total)
    assert torch.allclose(gi.compute().cpu(), ci.compute())


def test_graphed_update_matches_eager():
    """hipGraph-captured collection update == eager updates, incl. the
    device-side epoch protocol (logits-normalization) inside the graph."""
    torch.manual_seed(17)
    def make():
        return ma.MetricCollection({
            "acc": ma.MulticlassAccuracy(num_classes=50, average="macro", validate_args=False),
            "f1": ma.MulticlassF1Score(num_classes=50, average="weighted", validate_args=False),
            "confmat": ma.MulticlassConfusionMatrix(num_classes=50, validate_args=False),
            "exact": ma.MulticlassExactMatch(num_classes=50, validate_args=False),
            "auroc": ma.MulticlassAUROC(num_classes=50, thresholds=64, validate_args=False),
        }).to("cuda")

    batches = [(torch.randn(2048, 50, device="cuda", dtype=torch.bfloat16),
                torch.randint(0, 50, (2048,), device="cuda")) for _ in range(3)]
    eager = make()
    for p, t in batches:
        eager.update(p, t)
    res_eager = eager.compute()

    from metrics_amd.graphs import GraphedUpdate

    graphed_coll = make()
    graphed = GraphedUpdate(graphed_coll, batches[0][0], batches[0][1])
    for p, t in batches:
        graphed.update(p, t)
    res_graph = graphed_coll.compute()
    for k in res_eager:
        assert torch.allclose(res_eager[k].float(), res_graph[k].float(), atol=1e-6), k

    # reset_states + reuse
    graphed.reset_states()
    graphed.update(*batches[0])
    single = make()
    single.update(*batches[0])
    r1, r2 = graphed_coll.compute(), single.compute()
    for k in r1:
        assert torch.allclose(r1[k].float(), r2[k].float(), atol=1e-6), k


def test_graphed_update_rejects_list_states():
    from metrics_amd.graphs import GraphedUpdate

    m = ma.MulticlassPrecisionRecallCurve(num_classes=5, thresholds=None).to("cuda")
    with pytest.raises(RuntimeError, match="list state"):
        GraphedUpdate(m, torch.rand(8, 5, device="cuda"), torch.randint(0, 5, (8,), device="cuda"))


@pytest.mark.parametrize("average", ["micro", "macro", "weighted"])
def test_linear_stat_compute_matches_cpu(average):
    """The one-launch compute kernel must match the torch reduce chain."""
    torch.manual_seed(18)
    preds = torch.randn(4096, 37)
    target = torch.randint(0, 37, (4096,))
    target[1000:] = torch.randint(0, 20, (3096,))  # leave some classes empty
    makes = [
        lambda: ma.MulticlassAccuracy(num_classes=37, average=average),
        lambda: ma.MulticlassPrecision(num_classes=37, average=average),
        lambda: ma.MulticlassRecall(num_classes=37, average=average),
        lambda: ma.MulticlassF1Score(num_classes=37, average=average),
        lambda: ma.MulticlassFBetaScore(num_classes=37, beta=2.0, average=average),
        lambda: ma.MulticlassSpecificity(num_classes=37, average=average),
        lambda: ma.MulticlassNegativePredictiveValue(num_classes=37, average=average),
        lambda: ma.MulticlassHammingDistance(num_classes=37, average=average),
    ]
    for make in makes:
        g, c = make().to("cuda"), make()
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        rg, rc_ = g.compute(), c.compute()
        assert rg.ndim == rc_.ndim
        assert torch.allclose(rg.cpu(), rc_, atol=1e-6), (make().__class__.__name__, rg, rc_)


def test_linear_stat_compute_functional_gpu():
    from metrics_amd.functional import multiclass_precision

    torch.manual_seed(19)
    preds = torch.randn(1024, 10, device="cuda")
    target = torch.randint(0, 10, (1024,), device="cuda")
    v_gpu = multiclass_precision(preds, target, num_classes=10, average="macro")
    v_cpu = multiclass_precision(preds.cpu(), target.cpu(), num_classes=10, average="macro")
    assert torch.allclose(v_gpu.cpu(), v_cpu, atol=1e-6)


@pytest.mark.parametrize("average", ["macro", "weighted", "none"])
def test_auroc_ap_compute_kernel(average):
    torch.manual_seed(20)
    preds = torch.randn(4096, 29).softmax(-1)
    target = torch.randint(0, 29, (4096,))
    for make in (
        lambda: ma.MulticlassAUROC(num_classes=29, thresholds=75, average=average),
        lambda: ma.MulticlassAveragePrecision(num_classes=29, thresholds=75, average=average),
    ):
        g, c = make().to("cuda"), make()
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-5), make().__class__.__name__


@pytest.mark.parametrize("average", ["macro", "weighted"])
def test_multilabel_auroc_ap_compute_kernel(average):
    torch.manual_seed(21)
    preds = torch.rand(4096, 14)
    target = torch.randint(0, 2, (4096, 14))
    for make in (
        lambda: ma.MultilabelAUROC(num_labels=14, thresholds=60, average=average),
        lambda: ma.MultilabelAveragePrecision(num_labels=14, thresholds=60, average=average),
    ):
        g, c = make().to("cuda"), make()
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-5), make().__class__.__name__


def test_mcc_kappa_compute_fast_paths_gpu():
    torch.manual_seed(22)
    preds = torch.randn(4096, 100)
    target = torch.randint(0, 100, (4096,))
    for make in (
        lambda: ma.MulticlassMatthewsCorrCoef(num_classes=100),
        lambda: ma.MulticlassCohenKappa(num_classes=100),
        lambda: ma.MulticlassCohenKappa(num_classes=100, weights="quadratic"),
        lambda: ma.MulticlassCohenKappa(num_classes=100, weights="linear"),
        lambda: ma.MulticlassJaccardIndex(num_classes=100),
    ):
        g, c = make().to("cuda"), make()
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-5), make().__class__.__name__


def test_device_sweep_all_domains():
    """Representative metrics from every domain run update+compute on device."""
    torch.manual_seed(23)
    dev = "cuda"
    n = 256
    # classification
    ma.BinaryF1Score().to(dev)(torch.rand(n, device=dev), torch.randint(0, 2, (n,), device=dev))
    ma.MulticlassCalibrationError(num_classes=5).to(dev)(
        torch.randn(n, 5, device=dev).softmax(-1), torch.randint(0, 5, (n,), device=dev))
    ma.MultilabelRankingAveragePrecision(num_labels=6).to(dev)(
        torch.rand(n, 6, device=dev), torch.randint(0, 2, (n, 6), device=dev))
    # regression
    ma.PearsonCorrCoef().to(dev)(torch.randn(n, device=dev), torch.randn(n, device=dev))
    ma.SpearmanCorrCoef().to(dev)(torch.randn(n, device=dev), torch.randn(n, device=dev))
    ma.KendallRankCorrCoef().to(dev)(torch.randn(n, device=dev), torch.randn(n, device=dev))
    ma.R2Score().to(dev)(torch.randn(n, device=dev), torch.randn(n, device=dev))
    ma.MeanSquaredLogError().to(dev)(torch.rand(n, device=dev), torch.rand(n, device=dev))
    # retrieval
    idx = torch.randint(0, 8, (n,), device=dev)
    ma.RetrievalMRR().to(dev)(torch.rand(n, device=dev), torch.randint(0, 2, (n,), device=dev), indexes=idx)
    ma.RetrievalNormalizedDCG().to(dev)(torch.rand(n, device=dev), torch.randint(0, 2, (n,), device=dev), indexes=idx)
    # clustering
    ma.clustering.NormalizedMutualInfoScore().to(dev)(
        torch.randint(0, 5, (n,), device=dev), torch.randint(0, 5, (n,), device=dev))
    ma.clustering.CalinskiHarabaszScore().to(dev)(torch.randn(n, 4, device=dev), torch.randint(0, 3, (n,), device=dev))
    # nominal
    ma.CramersV(num_classes=5).to(dev)(torch.randint(0, 5, (n,), device=dev), torch.randint(0, 5, (n,), device=dev))
    # segmentation
    ma.segmentation.MeanIoU(num_classes=4, input_format="index").to(dev)(
        torch.randint(0, 4, (2, 16, 16), device=dev), torch.randint(0, 4, (2, 16, 16), device=dev))
    ma.segmentation.GeneralizedDiceScore(num_classes=4, input_format="index").to(dev)(
        torch.randint(0, 4, (2, 16, 16), device=dev), torch.randint(0, 4, (2, 16, 16), device=dev))
    # detection
    ma.detection.IntersectionOverUnion().to(dev)(
        [dict(boxes=torch.tensor([[0.0, 0, 10, 10]], device=dev), scores=torch.tensor([0.9], device=dev),
              labels=torch.tensor([0], device=dev))],
        [dict(boxes=torch.tensor([[2.0, 2, 12, 12]], device=dev), labels=torch.tensor([0], device=dev))])
    m = ma.detection.MeanAveragePrecision().to(dev)
    m.update(
        [dict(boxes=torch.tensor([[0.0, 0, 10, 10]], device=dev), scores=torch.tensor([0.9], device=dev),
              labels=torch.tensor([0], device=dev))],
        [dict(boxes=torch.tensor([[0.0, 0, 10, 10]], device=dev), labels=torch.tensor([0], device=dev))])
    assert abs(float(m.compute()["map"]) - 1.0) < 1e-6
    # image
    ma.StructuralSimilarityIndexMeasure(data_range=1.0).to(dev)(
        torch.rand(1, 3, 32, 32, device=dev), torch.rand(1, 3, 32, 32, device=dev))
    ma.PeakSignalNoiseRatio(data_range=1.0).to(dev)(
        torch.rand(1, 3, 32, 32, device=dev), torch.rand(1, 3, 32, 32, device=dev))
    # audio
    ma.audio.ScaleInvariantSignalDistortionRatio().to(dev)(
        torch.randn(2, 8000, device=dev), torch.randn(2, 8000, device=dev))
    ma.audio.SignalDistortionRatio().to(dev)(torch.randn(1, 4000, device=dev), torch.randn(1, 4000, device=dev))
    # aggregation + wrappers
    agg = ma.MeanMetric().to(dev)
    agg.update(torch.rand(n, device=dev))
    assert agg.compute().is_cuda
    boot = ma.wrappers.BootStrapper(ma.BinaryAccuracy(), num_bootstraps=4).to(dev)
    boot.update(torch.rand(n, device=dev), torch.randint(0, 2, (n,), device=dev))
    boot.compute()


def test_binary_auroc_ap_compute_kernel():
    torch.manual_seed(24)
    preds = torch.rand(20_000)
    target = torch.randint(0, 2, (20_000,))
    for make in (
        lambda: ma.BinaryAUROC(thresholds=120),
        lambda: ma.BinaryAveragePrecision(thresholds=120),
    ):
        g, c = make().to("cuda"), make()
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        rg, rc_ = g.compute(), c.compute()
        assert rg.ndim == 0
        assert torch.allclose(rg.cpu(), rc_, atol=1e-6), make().__class__.__name__
    # max_fpr path must still use the torch chain and agree with CPU
    g = ma.BinaryAUROC(thresholds=120, max_fpr=0.5).to("cuda")
    c = ma.BinaryAUROC(thresholds=120, max_fpr=0.5)
    g.update(preds.cuda(), target.cuda())
    c.update(preds, target)
    assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-6)


def test_fused_collection_update_plan():
    """Collection-level fusion: one kernel pass must equal separate updates."""
    torch.manual_seed(25)

    def make(device=None):
        c = ma.MetricCollection({
            "acc": ma.MulticlassAccuracy(num_classes=50, average="macro", validate_args=False),
            "prec": ma.MulticlassPrecision(num_classes=50, average="micro", validate_args=False),
            "kappa": ma.MulticlassCohenKappa(num_classes=50, validate_args=False),
            "confmat": ma.MulticlassConfusionMatrix(num_classes=50, validate_args=False),
            "exact": ma.MulticlassExactMatch(num_classes=50, validate_args=False),
        })
        return c.to(device) if device else c

    batches = [(torch.randn(2048, 50, dtype=torch.bfloat16), torch.randint(0, 50, (2048,))) for _ in range(3)]
    gpu, cpu = make("cuda"), make()
    for p, t in batches:
        gpu.update(p.cuda(), t.cuda())
        cpu.update(p, t)
    # the plan must actually engage
    assert gpu._fused_plan is not None and gpu._fused_plan is not False
    rg, rc_ = gpu.compute(), cpu.compute()
    for k in rg:
        assert torch.allclose(rg[k].float().cpu(), rc_[k].float(), atol=1e-6), k

    # with ignore_index on every participant
    gi = ma.MetricCollection({
        "acc": ma.MulticlassAccuracy(num_classes=50, ignore_index=-1, validate_args=False),
        "confmat": ma.MulticlassConfusionMatrix(num_classes=50, ignore_index=-1, validate_args=False),
        "exact": ma.MulticlassExactMatch(num_classes=50, ignore_index=-1, validate_args=False),
    }).to("cuda")
    ci = ma.MetricCollection({
        "acc": ma.MulticlassAccuracy(num_classes=50, ignore_index=-1, validate_args=False),
        "confmat": ma.MulticlassConfusionMatrix(num_classes=50, ignore_index=-1, validate_args=False),
        "exact": ma.MulticlassExactMatch(num_classes=50, ignore_index=-1, validate_args=False),
    })
    p = torch.randn(4096, 50)
    t = torch.randint(0, 50, (4096,))
    t[::7] = -1
    gi.update(p.cuda(), t.cuda())
    ci.update(p, t)
    r1, r2 = gi.compute(), ci.compute()
    for k in r1:
        assert torch.allclose(r1[k].float().cpu(), r2[k].float(), atol=1e-6), k


def test_fused_plan_not_built_with_validate_args():
    coll = ma.MetricCollection({
        "acc": ma.MulticlassAccuracy(num_classes=5),  # validate_args=True
        "confmat": ma.MulticlassConfusionMatrix(num_classes=5),
    }).to("cuda")
    coll.update(torch.randn(64, 5, device="cuda"), torch.randint(0, 5, (64,), device="cuda"))
    coll.update(torch.randn(64, 5, device="cuda"), torch.randint(0, 5, (64,), device="cuda"))
    assert coll._fused_plan is None  # validations would be skipped -> no fusion


@pytest.mark.parametrize("C", [3, 7, 63, 65, 1001, 4096])
def test_stat_kernel_boundary_class_counts(C):
    """Odd/unaligned C exercises the scalar load path; large C the vector path."""
    torch.manual_seed(26)
    B = 511
    preds = torch.randn(B, C)
    target = torch.randint(0, C, (B,))
    g = ma.MulticlassAccuracy(num_classes=C, average="micro").to("cuda")
    c = ma.MulticlassAccuracy(num_classes=C, average="micro")
    g.update(preds.cuda(), target.cuda())
    c.update(preds, target)
    assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-6), C


@pytest.mark.parametrize("T", [2, 5, 999, 4000])
def test_curve_kernel_boundary_thresholds(T):
    torch.manual_seed(27)
    preds = torch.rand(10_000)
    target = torch.randint(0, 2, (10_000,))
    g = ma.BinaryPrecisionRecallCurve(thresholds=T).to("cuda")
    c = ma.BinaryPrecisionRecallCurve(thresholds=T)
    g.update(preds.cuda(), target.cuda())
    c.update(preds, target)
    g._maybe_flush_lazy()
    assert torch.equal(g.confmat.cpu(), c.confmat), T


def test_curve_kernel_large_binary_input():
    torch.manual_seed(28)
    N = 4_000_000
    preds = torch.rand(N, device="cuda")
    target = torch.randint(0, 2, (N,), device="cuda")
    m = ma.BinaryAUROC(thresholds=500).to("cuda")
    m.update(preds, target)
    v = float(m.compute())
    assert 0.49 < v < 0.51  # random scores -> ~0.5


def test_mc_curve_nonuniform_thresholds():
    torch.manual_seed(29)
    thr = torch.tensor([0.01, 0.2, 0.21, 0.5, 0.93])  # non-uniform grid: binary-search path
    preds = torch.randn(2048, 9).softmax(-1)
    target = torch.randint(0, 9, (2048,))
    g = ma.MulticlassPrecisionRecallCurve(num_classes=9, thresholds=thr).to("cuda")
    c = ma.MulticlassPrecisionRecallCurve(num_classes=9, thresholds=thr)
    g.update(preds.cuda(), target.cuda())
    c.update(preds, target)
    g._maybe_flush_lazy()
    assert torch.equal(g.confmat.cpu(), c.confmat)


def test_tiny_batches_and_single_row():
    for B in (1, 2, 63):
        preds = torch.randn(B, 17)
        target = torch.randint(0, 17, (B,))
        g = ma.MulticlassF1Score(num_classes=17, average="macro").to("cuda")
        c = ma.MulticlassF1Score(num_classes=17, average="macro")
        g.update(preds.cuda(), target.cuda())
        c.update(preds, target)
        assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-6), B


def test_noncontiguous_inputs_gpu():
    """Kernels must handle (or contiguous-ify) transposed/strided inputs."""
    torch.manual_seed(30)
    base = torch.randn(40, 2048, device="cuda")  # transpose -> (2048, 40) non-contig
    preds = base.T
    assert not preds.is_contiguous()
    target = torch.randint(0, 40, (2048,), device="cuda")
    g = ma.MulticlassAccuracy(num_classes=40, average="macro").to("cuda")
    c = ma.MulticlassAccuracy(num_classes=40, average="macro")
    g.update(preds, target)
    c.update(preds.cpu(), target.cpu())
    assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-6)
    # strided binary preds
    bp = torch.rand(20_000, 2, device="cuda")[:, 0]
    bt = torch.randint(0, 2, (20_000,), device="cuda")
    gb = ma.BinaryF1Score().to("cuda")
    cb = ma.BinaryF1Score()
    gb.update(bp, bt)
    cb.update(bp.cpu(), bt.cpu())
    assert torch.allclose(gb.compute().cpu(), cb.compute(), atol=1e-6)


def test_retrieval_batched_gpu_matches_cpu():
    """Vectorized retrieval path on device == CPU reference (and the loop path)."""
    torch.manual_seed(31)
    n, q = 20_000, 1000
    idx = torch.randint(0, q, (n,))
    preds = torch.rand(n)
    target = torch.randint(0, 2, (n,))
    for cls, kw in (
        (ma.retrieval.RetrievalMAP, {}),
        (ma.retrieval.RetrievalMRR, {"top_k": 3}),
        (ma.retrieval.RetrievalNormalizedDCG, {"top_k": 10}),
        (ma.retrieval.RetrievalPrecision, {"top_k": 5}),
    ):
        g = cls(**kw).to("cuda")
        g.update(preds.cuda(), target.cuda(), indexes=idx.cuda())
        c = cls(**kw)
        c.update(preds, target, indexes=idx)
        assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-5), cls.__name__


def test_smallc_stat_kernel_gpu():
    """Thread-per-row small-C variant: confusion matrix + accuracy parity, C in {3, 48, 64, 100}."""
    torch.manual_seed(32)
    for C in (3, 48, 64, 100):
        preds = torch.randn(10_000, C, device="cuda", dtype=torch.bfloat16)
        target = torch.randint(0, C, (10_000,), device="cuda")
        g = ma.MulticlassConfusionMatrix(num_classes=C).to("cuda")
        g.update(preds, target)
        cm = g.compute().cpu()
        am = preds.float().argmax(1).cpu()
        ref = torch.zeros(C, C, dtype=torch.long)
        for t, p in zip(target.cpu().tolist(), am.tolist()):
            ref[t, p] += 1
        assert (cm.long() == ref).all(), C
        # label-path (no logits) small-C
        gl = ma.MulticlassAccuracy(num_classes=C, average="micro").to("cuda")
        gl.update(am.cuda(), target)
        cl = ma.MulticlassAccuracy(num_classes=C, average="micro")
        cl.update(am, target.cpu())
        assert torch.allclose(gl.compute().cpu(), cl.compute(), atol=1e-6), C


# ----------------------------------------------------------- K2 exact curve
def _torch_clf_curve(preds, target, weights=None, pos_label=1):
    """The (CPU-identical) torch formulation, used as the oracle for the
    rocPRIM sort+scan kernel path."""
    desc = torch.argsort(preds, descending=True)
    preds_s, target_s = preds[desc], target[desc]
    weight = weights[desc] if weights is not None else 1.0
    distinct = torch.where(preds_s[1:] - preds_s[:-1])[0]
    thr_idx = torch.nn.functional.pad(distinct, [0, 1], value=target_s.size(0) - 1)
    t = (target_s == pos_label).long()
    tps = torch.cumsum(t * weight, dim=0)[thr_idx]
    if weights is not None:
        fps = torch.cumsum((1 - t) * weight, dim=0)[thr_idx]
    else:
        fps = 1 + thr_idx - tps
    return fps, tps, preds_s[thr_idx]


@pytest.mark.parametrize("n", [1, 37, 5000, 200_000])
def test_hip_clf_curve_unweighted(n):
    torch.manual_seed(3)
    # heavy ties: scores quantized to 2 decimals
    preds = (torch.rand(n, device="cuda") * 100).round() / 100
    target = torch.randint(0, 2, (n,), device="cuda")
    fps, tps, thr = ops.hip_binary_clf_curve(preds, target)
    efps, etps, ethr = _torch_clf_curve(preds, target)
    assert torch.equal(thr, ethr.float())
    assert torch.equal(tps, etps.float())
    assert torch.equal(fps, efps.float())


def test_hip_clf_curve_weighted():
    torch.manual_seed(4)
    n = 10_000
    preds = (torch.rand(n, device="cuda") * 50).round() / 50
    target = torch.randint(0, 2, (n,), device="cuda")
    w = torch.rand(n, device="cuda")
    fps, tps, thr = ops.hip_binary_clf_curve(preds, target, w)
    efps, etps, ethr = _torch_clf_curve(preds, target, w)
    assert torch.equal(thr, ethr)
    assert torch.allclose(tps, etps, atol=1e-3)
    assert torch.allclose(fps, efps, atol=1e-3)


def test_hip_clf_curve_bf16_and_poslabel():
    torch.manual_seed(5)
    n = 4096
    preds = torch.rand(n, device="cuda", dtype=torch.bfloat16)
    target = torch.randint(0, 2, (n,), device="cuda")
    fps, tps, thr = ops.hip_binary_clf_curve(preds, target, pos_label=0)
    efps, etps, ethr = _torch_clf_curve(preds.float(), target, pos_label=0)
    assert torch.equal(tps, etps.float()) and torch.equal(fps, efps.float())
    assert torch.equal(thr, ethr)


def test_exact_roc_auroc_end_to_end_gpu():
    """thresholds=None BinaryROC/AUROC/PrecisionRecallCurve run the HIP K2
    path on GPU and must match the CPU (torch sort) result."""
    torch.manual_seed(6)
    n = 50_000
    preds = torch.rand(n)
    target = torch.randint(0, 2, (n,))
    for cls in (ma.BinaryROC, ma.BinaryPrecisionRecallCurve):
        mg = cls(thresholds=None).to("cuda")
        mg.update(preds.cuda(), target.cuda())
        res_g = mg.compute()
        mc = cls(thresholds=None)
        mc.update(preds, target)
        res_c = mc.compute()
        for a, b in zip(res_g, res_c):
            assert torch.allclose(a.cpu(), b, atol=1e-6), cls.__name__
    ag = ma.BinaryAUROC(thresholds=None).to("cuda")
    ag.update(preds.cuda(), target.cuda())
    ac = ma.BinaryAUROC(thresholds=None)
    ac.update(preds, target)
    assert torch.allclose(ag.compute().cpu(), ac.compute(), atol=1e-6)


def test_exact_multiclass_curve_gpu():
    torch.manual_seed(7)
    preds = torch.randn(2000, 7).softmax(-1)
    target = torch.randint(0, 7, (2000,))
    mg = ma.MulticlassAveragePrecision(num_classes=7, thresholds=None, average="macro").to("cuda")
    mg.update(preds.cuda(), target.cuda())
    mc = ma.MulticlassAveragePrecision(num_classes=7, thresholds=None, average="macro")
    mc.update(preds, target)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-6)


# ----------------------------------------------------------- K8 fused SSIM
@pytest.mark.parametrize("kwargs", [
    {},
    {"sigma": 1.0},
    {"gaussian_kernel": False, "kernel_size": 7},
    {"data_range": 2.5},
    {"data_range": (0.1, 0.9)},
    {"reduction": "none"},
])
def test_ssim_fused_vs_cpu(kwargs):
    from metrics_amd.functional.image import structural_similarity_index_measure as ssim

    torch.manual_seed(31)
    p = torch.rand(3, 2, 57, 83)
    t = torch.rand(3, 2, 57, 83)
    got = ssim(p.cuda(), t.cuda(), **kwargs)
    exp = ssim(p, t, **kwargs)
    assert torch.allclose(got.cpu(), exp, atol=1e-5), (got, exp)


def test_ssim_fused_contrast_sensitivity_and_ms():
    from metrics_amd.functional.image import (
        multiscale_structural_similarity_index_measure as ms_ssim,
        structural_similarity_index_measure as ssim,
    )

    torch.manual_seed(32)
    p = torch.rand(2, 3, 200, 200)
    t = torch.rand(2, 3, 200, 200)
    g_sim, g_cs = ssim(p.cuda(), t.cuda(), return_contrast_sensitivity=True)
    c_sim, c_cs = ssim(p, t, return_contrast_sensitivity=True)
    assert torch.allclose(g_sim.cpu(), c_sim, atol=1e-5)
    assert torch.allclose(g_cs.cpu(), c_cs, atol=1e-5)
    assert torch.allclose(ms_ssim(p.cuda(), t.cuda()).cpu(), ms_ssim(p, t), atol=1e-4)


def test_ssim_fused_bf16():
    from metrics_amd.functional.image import structural_similarity_index_measure as ssim

    torch.manual_seed(33)
    p = torch.rand(2, 1, 64, 64)
    t = torch.rand(2, 1, 64, 64)
    got = ssim(p.cuda().bfloat16(), t.cuda().bfloat16(), data_range=1.0)
    exp = ssim(p, t, data_range=1.0)
    assert torch.allclose(got.float().cpu(), exp, atol=2e-2)


def test_ssim_metric_gpu_end_to_end():
    torch.manual_seed(34)
    p = torch.rand(4, 3, 96, 96)
    t = torch.rand(4, 3, 96, 96)
    mg = ma.StructuralSimilarityIndexMeasure(data_range=1.0).to("cuda")
    mg.update(p.cuda(), t.cuda())
    mc = ma.StructuralSimilarityIndexMeasure(data_range=1.0)
    mc.update(p, t)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-5)


# ------------------------------------------------------------- K8 erosion
def test_binary_erosion_gpu_vs_cpu():
    from metrics_amd.functional.segmentation.utils import binary_erosion, generate_binary_structure

    torch.manual_seed(35)
    img = (torch.rand(2, 3, 41, 53) > 0.4).int()
    for structure, origin in [
        (None, None),
        (torch.ones(3, 3, dtype=torch.int32), None),
        (torch.tensor([[0, 1, 0], [1, 1, 1], [0, 1, 0]], dtype=torch.int32), None),
        (torch.ones(5, 3, dtype=torch.int32), (2, 1)),
        (torch.tensor([[1, 1], [0, 1]], dtype=torch.int32), (0, 0)),
    ]:
        cpu = binary_erosion(img, structure, origin)
        gpu = binary_erosion(
            img.cuda(), structure.cuda() if structure is not None else None, origin
        )
        assert torch.equal(gpu.cpu(), cpu), (structure, origin)
    # border_value=1
    cpu = binary_erosion(img, border_value=1)
    gpu = binary_erosion(img.cuda(), border_value=1)
    assert torch.equal(gpu.cpu(), cpu)


def test_hausdorff_gpu_vs_cpu():
    torch.manual_seed(36)
    p = torch.randint(0, 2, (2, 2, 32, 32))
    t = torch.randint(0, 2, (2, 2, 32, 32))
    mg = ma.segmentation.HausdorffDistance(num_classes=2).to("cuda")
    mg.update(p.cuda(), t.cuda())
    mc = ma.segmentation.HausdorffDistance(num_classes=2)
    mc.update(p, t)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-5)


# ---------------------------------------------------------- K5 calibration
@pytest.mark.parametrize("n_bins", [10, 15, 99])
@pytest.mark.parametrize("norm", ["l1", "l2", "max"])
def test_calibration_error_gpu_vs_cpu(n_bins, norm):
    torch.manual_seed(41)
    preds = torch.rand(5000)
    target = torch.randint(0, 2, (5000,))
    mg = ma.BinaryCalibrationError(n_bins=n_bins, norm=norm).to("cuda")
    mg.update(preds.cuda(), target.cuda())
    mc = ma.BinaryCalibrationError(n_bins=n_bins, norm=norm)
    mc.update(preds, target)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-5)


def test_multiclass_calibration_error_gpu():
    torch.manual_seed(42)
    preds = torch.randn(2000, 9).softmax(-1)
    target = torch.randint(0, 9, (2000,))
    mg = ma.MulticlassCalibrationError(num_classes=9).to("cuda")
    mg.update(preds.cuda(), target.cuda())
    mc = ma.MulticlassCalibrationError(num_classes=9)
    mc.update(preds, target)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-5)


def test_calib_bins_deterministic_repeat():
    torch.manual_seed(43)
    conf = torch.rand(200_000, device="cuda")
    acc = (torch.rand(200_000, device="cuda") > 0.5).float()
    bounds = torch.linspace(0, 1, 16, device="cuda")
    from metrics_amd.ops import _hip

    a1 = _hip.calib_bins(conf, acc, bounds)
    a2 = _hip.calib_bins(conf, acc, bounds)
    for x, y in zip(a1, a2):
        assert torch.equal(x, y)


# --------------------------------------------------------------- K3 top-k
@pytest.mark.parametrize("k", [2, 3, 5])
@pytest.mark.parametrize("ignore_index", [None, 2, -1])
@pytest.mark.parametrize("average", ["micro", "macro"])
def test_topk_stat_gpu_vs_cpu(k, ignore_index, average):
    torch.manual_seed(44)
    C = 11
    preds = torch.randn(3000, C).softmax(-1)
    target = torch.randint(0, C, (3000,))
    if ignore_index is not None:
        target[torch.rand(3000) < 0.1] = ignore_index
    args = dict(num_classes=C, top_k=k, average=average, ignore_index=ignore_index, validate_args=False)
    mg = ma.MulticlassAccuracy(**args).to("cuda")
    mg.update(preds.cuda(), target.cuda())
    mc = ma.MulticlassAccuracy(**args)
    mc.update(preds, target)
    assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-6)
    # stat scores directly
    sg = ma.MulticlassStatScores(**{**args, "average": None}).to("cuda")
    sg.update(preds.cuda(), target.cuda())
    sc = ma.MulticlassStatScores(**{**args, "average": None})
    sc.update(preds, target)
    assert torch.equal(sg.compute().cpu(), sc.compute())


def test_topk_precision_recall_gpu():
    torch.manual_seed(45)
    C = 7
    preds = torch.randn(1000, C).softmax(-1)
    target = torch.randint(0, C, (1000,))
    for cls in (ma.MulticlassPrecision, ma.MulticlassRecall, ma.MulticlassF1Score):
        mg = cls(num_classes=C, top_k=3, average="macro").to("cuda")
        mg.update(preds.cuda(), target.cuda())
        mc = cls(num_classes=C, top_k=3, average="macro")
        mc.update(preds, target)
        assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-6), cls.__name__


# -------------------------------------------------- host-sync regression
def test_update_has_no_device_host_sync():
    """The hot update path must never synchronize with the host (the
    reference guards this with torch.cuda.set_sync_debug_mode — same idea:
    tests/unittests/classification/test_accuracy.py:352-408)."""
    kw = dict(num_classes=50, validate_args=False)
    coll = ma.MetricCollection({
        "acc": ma.MulticlassAccuracy(average="micro", **kw),
        "f1": ma.MulticlassF1Score(average="macro", **kw),
        "confmat": ma.MulticlassConfusionMatrix(**kw),
        "auroc": ma.MulticlassAUROC(average="macro", thresholds=20, **kw),
    }).to("cuda")
    preds = torch.randn(256, 50, device="cuda")
    target = torch.randint(0, 50, (256,), device="cuda")
    coll.update(preds, target)  # warmup (pools, uniform-threshold cache)
    torch.cuda.synchronize()
    torch.cuda.set_sync_debug_mode("error")
    try:
        coll.update(preds, target)
        coll.update(preds, target)
    finally:
        torch.cuda.set_sync_debug_mode("default")
    torch.cuda.synchronize()


def test_sync_debug_mode_actually_fires():
    """Negative control: a known-syncing op must raise under the mode."""
    t = torch.rand(10, device="cuda")
    torch.cuda.set_sync_debug_mode("error")
    try:
        with pytest.raises(RuntimeError):
            t.item()
    finally:
        torch.cuda.set_sync_debug_mode("default")


def test_retrieval_sort_kernel_vs_torch():
    from metrics_amd.ops import _hip

    torch.manual_seed(71)
    n = 100_000
    idx = torch.randint(0, 5000, (n,), device="cuda")
    preds = torch.rand(n, device="cuda")
    preds[::97] = preds[0]  # inject ties
    order, by_index = _hip.retrieval_sort(idx, preds)
    o1 = torch.argsort(preds, descending=True, stable=True)
    o2 = torch.argsort(idx[o1], stable=True)
    ref_order = o1[o2]
    ref_by_index = torch.argsort(idx, stable=True)
    assert torch.equal(order, ref_order)
    assert torch.equal(by_index, ref_by_index)


def test_retrieval_metric_kernel_path_gpu():
    torch.manual_seed(72)
    idx = torch.randint(0, 500, (20_000,))
    preds = torch.rand(20_000)
    target = torch.randint(0, 2, (20_000,))
    for cls, kw in ((ma.retrieval.RetrievalMAP, {}), (ma.retrieval.RetrievalNormalizedDCG, {}),
                    (ma.retrieval.RetrievalPrecision, {"top_k": 5})):
        mg = cls(**kw).to("cuda")
        mg.update(preds.cuda(), target.cuda(), indexes=idx.cuda())
        mc = cls(**kw)
        mc.update(preds, target, indexes=idx)
        assert torch.allclose(mg.compute().cpu(), mc.compute(), atol=1e-6), cls.__name__


def test_mc_clf_curve_batched_vs_per_class():
    """One composite-key sort must reproduce the per-class K2 curves exactly
    (multiclass AND multilabel), including heavy score ties."""
    from metrics_amd.ops import _hip

    torch.manual_seed(81)
    B, C = 3000, 9
    probs = ((torch.rand(B, C, device="cuda") * 50).round() / 50)
    target = torch.randint(0, C, (B,), device="cuda")
    batched = _hip.mc_clf_curve(probs, target, multilabel=False)
    for c in range(C):
        f, t, th = _hip.binary_clf_curve(probs[:, c].contiguous(), target, pos_label=c)
        bf, bt, bth, n_pos, n_neg = batched[c]
        assert torch.equal(bf, f) and torch.equal(bt, t) and torch.equal(bth, th), c
        assert n_pos == int((target == c).sum()) and n_pos + n_neg == B
    # multilabel
    ml_t = torch.randint(0, 2, (B, C), device="cuda")
    batched_ml = _hip.mc_clf_curve(probs, ml_t, multilabel=True)
    for c in range(C):
        f, t, th = _hip.binary_clf_curve(probs[:, c].contiguous(), ml_t[:, c].contiguous(), pos_label=1)
        bf, bt, bth, n_pos, n_neg = batched_ml[c]
        assert torch.equal(bf, f) and torch.equal(bt, t) and torch.equal(bth, th), c
        assert n_pos == int(ml_t[:, c].sum()) and n_pos + n_neg == B


def test_exact_multiclass_roc_prc_gpu_batched():
    torch.manual_seed(82)
    preds = torch.randn(1500, 6).softmax(-1)
    target = torch.randint(0, 6, (1500,))
    for cls, kw in [
        (ma.MulticlassROC, {"num_classes": 6, "thresholds": None}),
        (ma.MulticlassPrecisionRecallCurve, {"num_classes": 6, "thresholds": None}),
        (ma.MultilabelROC, {"num_labels": 6, "thresholds": None}),
    ]:
        if "num_labels" in kw:
            tgt = torch.randint(0, 2, (1500, 6))
        else:
            tgt = target
        mg = cls(**kw).to("cuda")
        mg.update(preds.cuda(), tgt.cuda())
        mc = cls(**kw)
        mc.update(preds, tgt)
        rg, rc_ = mg.compute(), mc.compute()
        for a, b in zip(rg, rc_):
            if isinstance(a, list):
                for x, y in zip(a, b):
                    assert torch.allclose(x.cpu(), y, atol=1e-6), cls.__name__
            else:
                assert torch.allclose(a.cpu(), b, atol=1e-6), cls.__name__


def test_lazy_curve_confmat_accumulation():
    """Lazy curve path: histograms accumulate across updates; the suffix/
    confmat materialization defers to the first state read and matches the
    per-update (eager) result exactly."""
    import metrics_amd as ma

    torch.manual_seed(11)
    batches = [
        (torch.randn(256, 10, device="cuda"), torch.randint(0, 10, (256,), device="cuda"))
        for _ in range(4)
    ]
    lazy = ma.MulticlassAUROC(num_classes=10, thresholds=50).to("cuda")
    for p, t in batches:
        lazy.update(p, t)
    # histogram accumulated, confmat still all-zero until a state read
    assert lazy.__dict__.get("_lazy_dirty") is True
    assert int(lazy.confmat.sum().item()) == 0
    out_lazy = lazy.compute()
    assert lazy.__dict__.get("_lazy_dirty") is False

    # eager reference: flush after every update
    eager = ma.MulticlassAUROC(num_classes=10, thresholds=50).to("cuda")
    for p, t in batches:
        eager.update(p, t)
        eager._maybe_flush_lazy()
    assert torch.equal(lazy.confmat, eager.confmat)
    assert torch.allclose(out_lazy, eager.compute())

    # CPU oracle
    cpu = ma.MulticlassAUROC(num_classes=10, thresholds=50)
    for p, t in batches:
        cpu.update(p.cpu().float(), t.cpu())
    assert torch.allclose(out_lazy.cpu(), cpu.compute(), atol=1e-5)


def test_lazy_curve_state_dict_and_reset():
    import metrics_amd as ma

    torch.manual_seed(12)
    p = torch.randn(128, 5, device="cuda")
    t = torch.randint(0, 5, (128,), device="cuda")
    m = ma.MulticlassPrecisionRecallCurve(num_classes=5, thresholds=20).to("cuda")
    m.update(p, t)
    m.state_dict()  # any state access must flush (confmat itself is a
    # non-persistent state, like the reference's, so it is not SAVED)
    assert m.__dict__.get("_lazy_dirty") is False
    assert int(m.confmat.sum().item()) > 0
    pr1 = m.compute()
    # reset drops pending histogram counts
    m3 = ma.MulticlassPrecisionRecallCurve(num_classes=5, thresholds=20).to("cuda")
    m3.update(p, t)
    m3.reset()
    m3.update(p, t)
    pr3 = m3.compute()
    for a, b in zip(pr1, pr3):
        assert torch.equal(a, b)
    # device moves flush first (the kernel buffers stay device-paired)
    m4 = ma.MulticlassPrecisionRecallCurve(num_classes=5, thresholds=20).to("cuda")
    m4.update(p, t)
    m4 = m4.cpu()
    pr4 = m4.compute()
    for a, b in zip(pr1, pr4):
        assert torch.equal(a.cpu(), b)


def test_lazy_curve_forward_matches_reference_semantics():
    import metrics_amd as ma

    torch.manual_seed(13)
    m = ma.MulticlassAUROC(num_classes=5, thresholds=25).to("cuda")
    cpu = ma.MulticlassAUROC(num_classes=5, thresholds=25)
    for i in range(3):
        p = torch.randn(64, 5, device="cuda")
        t = torch.randint(0, 5, (64,), device="cuda")
        v_gpu = m(p, t)
        v_cpu = cpu(p.cpu().float(), t.cpu())
        assert torch.allclose(v_gpu.cpu(), v_cpu, atol=1e-5), i
    assert torch.allclose(m.compute().cpu(), cpu.compute(), atol=1e-5)


def test_lazy_curve_collection_fused_and_groups():
    """The fused collection path and compute groups stay correct with lazy
    curve state (members alias the leader's confmat; flush happens before
    any member computes)."""
    import metrics_amd as ma

    torch.manual_seed(14)
    coll = ma.MetricCollection(
        {
            "acc": ma.MulticlassAccuracy(num_classes=8, average="micro"),
            "auroc": ma.MulticlassAUROC(num_classes=8, thresholds=40),
            "ap": ma.MulticlassAveragePrecision(num_classes=8, thresholds=40),
        }
    ).to("cuda")
    cpu = ma.MetricCollection(
        {
            "acc": ma.MulticlassAccuracy(num_classes=8, average="micro"),
            "auroc": ma.MulticlassAUROC(num_classes=8, thresholds=40),
            "ap": ma.MulticlassAveragePrecision(num_classes=8, thresholds=40),
        }
    )
    for i in range(3):
        p = torch.randn(256, 8, device="cuda")
        t = torch.randint(0, 8, (256,), device="cuda")
        coll.update(p, t)
        cpu.update(p.cpu().float(), t.cpu())
    out = coll.compute()
    ref = cpu.compute()
    for k in ref:
        assert torch.allclose(out[k].cpu(), ref[k], atol=1e-5), k


def test_confmat_scalars_fused_vs_cpu():
    """Fused MCC/kappa/jaccard scalars match the CPU torch chains, and the
    version cache invalidates across kernel-side confmat updates."""
    import metrics_amd as ma

    torch.manual_seed(21)
    mcc_g = ma.MulticlassMatthewsCorrCoef(num_classes=37).to("cuda")
    kap_g = ma.MulticlassCohenKappa(num_classes=37).to("cuda")
    jac_g = ma.MulticlassJaccardIndex(num_classes=37, average="macro").to("cuda")
    mcc_c = ma.MulticlassMatthewsCorrCoef(num_classes=37)
    kap_c = ma.MulticlassCohenKappa(num_classes=37)
    jac_c = ma.MulticlassJaccardIndex(num_classes=37, average="macro")
    for step in range(3):
        p = torch.randn(512, 37, device="cuda").softmax(-1)
        t = torch.randint(0, 37, (512,), device="cuda")
        for m in (mcc_g, kap_g, jac_g):
            m.update(p, t)
        for m in (mcc_c, kap_c, jac_c):
            m.update(p.cpu(), t.cpu())
        # compute EVERY step: a stale cache would freeze the value at step 0
        assert torch.allclose(mcc_g.compute().cpu(), mcc_c.compute(), atol=1e-5), step
        assert torch.allclose(kap_g.compute().cpu(), kap_c.compute(), atol=1e-5), step
        assert torch.allclose(jac_g.compute().cpu(), jac_c.compute(), atol=1e-5), step

    # skewed degenerate case: single predicted class
    m1 = ma.MulticlassMatthewsCorrCoef(num_classes=5).to("cuda")
    m2 = ma.MulticlassMatthewsCorrCoef(num_classes=5)
    p = torch.zeros(64, 5, device="cuda")
    p[:, 2] = 10.0
    t = torch.randint(0, 5, (64,), device="cuda")
    m1.update(p, t)
    m2.update(p.cpu(), t.cpu())
    assert torch.allclose(m1.compute().cpu(), m2.compute(), atol=1e-6)


def test_confmat_scalars_in_collection():
    """The three confmat-derived metrics share one compute-group confmat: the
    fused scalars are computed once per generation and stay correct."""
    import metrics_amd as ma

    torch.manual_seed(22)
    coll = ma.MetricCollection(
        {
            "mcc": ma.MulticlassMatthewsCorrCoef(num_classes=19),
            "kappa": ma.MulticlassCohenKappa(num_classes=19),
            "jaccard": ma.MulticlassJaccardIndex(num_classes=19),
            "confmat": ma.MulticlassConfusionMatrix(num_classes=19),
        }
    ).to("cuda")
    ref = ma.MetricCollection(
        {
            "mcc": ma.MulticlassMatthewsCorrCoef(num_classes=19),
            "kappa": ma.MulticlassCohenKappa(num_classes=19),
            "jaccard": ma.MulticlassJaccardIndex(num_classes=19),
            "confmat": ma.MulticlassConfusionMatrix(num_classes=19),
        }
    )
    for _ in range(2):
        p = torch.randn(300, 19, device="cuda").softmax(-1)
        t = torch.randint(0, 19, (300,), device="cuda")
        coll.update(p, t)
        ref.update(p.cpu(), t.cpu())
        out, exp = coll.compute(), ref.compute()
        for k in exp:
            assert torch.allclose(out[k].float().cpu(), exp[k].float(), atol=1e-5), k


@pytest.mark.parametrize("average", ["macro", "weighted", "none"])
@pytest.mark.parametrize("cls_name", ["MulticlassAUROC", "MulticlassAveragePrecision"])
def test_curve_auc_average_variants_gpu(cls_name, average):
    """All fused-path average modes of the curve-area metrics match CPU."""
    import metrics_amd as ma

    torch.manual_seed(31)
    cls = getattr(ma, cls_name)
    g = cls(num_classes=13, thresholds=60, average=average).to("cuda")
    c = cls(num_classes=13, thresholds=60, average=average)
    for _ in range(2):
        p = torch.randn(300, 13, device="cuda").softmax(-1)
        t = torch.randint(0, 13, (300,), device="cuda")
        g.update(p, t)
        c.update(p.cpu(), t.cpu())
    assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-5)


@pytest.mark.parametrize("aggregation", ["mean", "median", "min", "max"])
def test_retrieval_aggregation_gpu(aggregation):
    """Aggregation modes ride the batched GPU scoring path (composite-key
    rocPRIM sort) and match CPU."""
    import metrics_amd as ma

    torch.manual_seed(41)
    p = torch.rand(2000, device="cuda")
    t = torch.randint(0, 2, (2000,), device="cuda")
    idx = torch.randint(0, 50, (2000,), device="cuda")
    g = ma.retrieval.RetrievalMAP(aggregation=aggregation).to("cuda")
    c = ma.retrieval.RetrievalMAP(aggregation=aggregation)
    g.update(p, t, indexes=idx)
    c.update(p.cpu(), t.cpu(), indexes=idx.cpu())
    assert torch.allclose(g.compute().cpu(), c.compute(), atol=1e-6)
