"""MetricTester-style randomized oracle sweep (reference
tests/unittests/_helpers/testers.py:85-250 coverage model).

Every case runs the full check battery: pickle/clone, batch-by-batch
``forward`` vs the oracle, accumulated ``compute`` vs the oracle on all
data, reset-to-defaults — and, per case, a 2-process gloo DDP variant
(interleaved batches, synced compute vs oracle on ALL data) plus a
double-dtype consistency run. sklearn/scipy are the oracles wherever they
model the semantics; top_k>1 uses an independent numpy implementation
(sklearn has no per-class top-k).
"""
import numpy as np
import pytest
import torch
from sklearn import metrics as skm

import metrics_amd as ma
from tests.unittests._helpers import (
    run_class_metric_ddp_test,
    run_class_metric_test,
    run_dtype_test,
    seed_all,
)

C = 7
L = 4
B = 29  # deliberately odd


def _mc_inputs(seed, kind="logits", ignore_index=None):
    seed_all(seed)
    if kind == "logits":
        preds = torch.randn(4, B, C)
    elif kind == "probs":
        preds = torch.randn(4, B, C).softmax(-1)
    else:  # labels
        preds = torch.randint(0, C, (4, B))
    target = torch.randint(0, C, (4, B))
    if ignore_index is not None:
        target[torch.rand(4, B) < 0.15] = ignore_index
    return preds, target


def _pred_labels(p):
    return p.argmax(-1).numpy() if p.is_floating_point() else p.numpy()


def _filtered(p, t, ignore_index):
    tl = t.numpy().ravel()
    pl = _pred_labels(p).ravel()
    if ignore_index is not None:
        keep = tl != ignore_index
        tl, pl = tl[keep], pl[keep]
    return pl, tl


# --------------------------------------------------------------- multiclass
_MC_PRF = [
    (ma.MulticlassPrecision, skm.precision_score),
    (ma.MulticlassRecall, skm.recall_score),
    (ma.MulticlassF1Score, skm.f1_score),
]


@pytest.mark.parametrize("average", ["micro", "macro", "weighted", None])
@pytest.mark.parametrize("ignore_index", [None, 3])
@pytest.mark.parametrize("kind", ["logits", "probs", "labels"])
@pytest.mark.parametrize(("cls", "sk_fn"), _MC_PRF)
def test_mc_prf_sweep(cls, sk_fn, kind, ignore_index, average):
    preds, target = _mc_inputs(11, kind, ignore_index)

    def ref(p, t):
        pl, tl = _filtered(p, t, ignore_index)
        return sk_fn(tl, pl, labels=range(C), average=average, zero_division=0)

    args = {"num_classes": C, "average": average, "ignore_index": ignore_index}
    run_class_metric_test(cls, ref, preds, target, args)
    run_class_metric_ddp_test(cls, ref, preds, target, args)


@pytest.mark.parametrize("ignore_index", [None, 0])
@pytest.mark.parametrize("average", ["micro", "macro"])
def test_mc_accuracy_sweep(average, ignore_index):
    preds, target = _mc_inputs(12, "logits", ignore_index)

    def ref(p, t):
        pl, tl = _filtered(p, t, ignore_index)
        if average == "micro":
            return skm.accuracy_score(tl, pl)
        return skm.recall_score(tl, pl, labels=range(C), average="macro", zero_division=0)

    args = {"num_classes": C, "average": average, "ignore_index": ignore_index}
    run_class_metric_test(ma.MulticlassAccuracy, ref, preds, target, args)
    run_class_metric_ddp_test(ma.MulticlassAccuracy, ref, preds, target, args)
    run_dtype_test(ma.MulticlassAccuracy, preds, target, args)


def _np_topk_recall(probs, t, k, average, ignore_index=None):
    """Independent top-k per-class oracle: tp_c = |t==c and c in topk|."""
    probs = probs.numpy() if isinstance(probs, torch.Tensor) else probs
    t = t.numpy() if isinstance(t, torch.Tensor) else t
    probs = probs.reshape(-1, probs.shape[-1])
    t = t.ravel()
    if ignore_index is not None:
        keep = t != ignore_index
        probs, t = probs[keep], t[keep]
    topk = np.argsort(-probs, axis=1, kind="stable")[:, :k]
    hit = (topk == t[:, None]).any(1)
    tp = np.bincount(t[hit], minlength=C).astype(float)
    support = np.bincount(t, minlength=C).astype(float)
    if average == "micro":
        return tp.sum() / support.sum()
    per_class = np.divide(tp, support, out=np.zeros_like(tp), where=support > 0)
    if average == "macro":
        return per_class[support > 0].mean()
    return per_class  # none


@pytest.mark.parametrize("ignore_index", [None, 2])
@pytest.mark.parametrize("average", ["micro", "macro", None])
@pytest.mark.parametrize("top_k", [2, 3])
def test_mc_topk_accuracy_sweep(top_k, average, ignore_index):
    preds, target = _mc_inputs(13, "probs", ignore_index)

    def ref(p, t):
        return _np_topk_recall(p, t, top_k, average, ignore_index)

    args = {"num_classes": C, "average": average, "top_k": top_k, "ignore_index": ignore_index}
    run_class_metric_test(ma.MulticlassAccuracy, ref, preds, target, args)
    run_class_metric_ddp_test(ma.MulticlassAccuracy, ref, preds, target, args)


@pytest.mark.parametrize("ignore_index", [None, 1])
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.MulticlassCohenKappa, skm.cohen_kappa_score),
        (ma.MulticlassMatthewsCorrCoef, skm.matthews_corrcoef),
    ],
)
def test_mc_agreement_sweep(cls, sk_fn, ignore_index):
    preds, target = _mc_inputs(14, "logits", ignore_index)

    def ref(p, t):
        pl, tl = _filtered(p, t, ignore_index)
        return sk_fn(tl, pl)

    args = {"num_classes": C, "ignore_index": ignore_index}
    run_class_metric_test(cls, ref, preds, target, args)
    run_class_metric_ddp_test(cls, ref, preds, target, args)


def test_mc_samplewise_multidim():
    seed_all(15)
    D = 11
    preds = torch.randn(4, B, C, D)
    target = torch.randint(0, C, (4, B, D))

    def ref(p, t):
        pl = p.argmax(-2).numpy()  # (B, D)
        tl = t.numpy()
        return np.array([skm.accuracy_score(tl[i].ravel(), pl[i].ravel()) for i in range(tl.shape[0])])

    args = {"num_classes": C, "average": "micro", "multidim_average": "samplewise"}
    # samplewise => cat state; batch values ARE per-sample, compute concatenates
    m = ma.MulticlassAccuracy(**args)
    for i in range(4):
        batch_val = m(preds[i], target[i])
        assert np.allclose(batch_val.numpy(), ref(preds[i], target[i]), atol=1e-6)
    total = m.compute()
    assert total.shape == (4 * B,)
    expected = np.concatenate([ref(preds[i], target[i]) for i in range(4)])
    assert np.allclose(total.numpy(), expected, atol=1e-6)


@pytest.mark.parametrize("average", ["macro", "weighted"])
def test_mc_auroc_ap_exact_sweep(average):
    seed_all(16)
    preds = torch.randn(4, B, C).softmax(-1)
    target = torch.randint(0, C, (4, B))

    def ref_auroc(p, t):
        return skm.roc_auc_score(
            t.numpy().ravel(), p.reshape(-1, C).numpy(), multi_class="ovr",
            average=average, labels=range(C),
        )

    args = {"num_classes": C, "thresholds": None, "average": average}
    run_class_metric_test(ma.MulticlassAUROC, ref_auroc, preds, target, args, check_batch=False)
    run_class_metric_ddp_test(ma.MulticlassAUROC, ref_auroc, preds, target, args)

    def ref_ap(p, t):
        tl = t.numpy().ravel()
        pp = p.reshape(-1, C).numpy()
        onehot = np.eye(C)[tl]
        scores = [skm.average_precision_score(onehot[:, c], pp[:, c]) for c in range(C)]
        scores = np.nan_to_num(np.array(scores), nan=0.0)
        if average == "macro":
            present = np.bincount(tl, minlength=C) > 0
            return scores[present].mean()
        w = np.bincount(tl, minlength=C) / len(tl)
        return (scores * w).sum()

    run_class_metric_test(
        ma.MulticlassAveragePrecision, ref_ap, preds, target, args, check_batch=False
    )


# ------------------------------------------------------------------- binary
@pytest.mark.parametrize("ignore_index", [None, -1])
@pytest.mark.parametrize("kind", ["probs", "logits"])
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.BinaryAccuracy, skm.accuracy_score),
        (ma.BinaryPrecision, lambda t, p: skm.precision_score(t, p, zero_division=0)),
        (ma.BinaryRecall, lambda t, p: skm.recall_score(t, p, zero_division=0)),
        (ma.BinaryF1Score, lambda t, p: skm.f1_score(t, p, zero_division=0)),
        (ma.BinaryMatthewsCorrCoef, skm.matthews_corrcoef),
        (ma.BinaryCohenKappa, skm.cohen_kappa_score),
    ],
)
def test_binary_sweep(cls, sk_fn, kind, ignore_index):
    seed_all(17)
    preds = torch.rand(4, B) if kind == "probs" else torch.randn(4, B) * 2
    target = torch.randint(0, 2, (4, B))
    if ignore_index is not None:
        target[torch.rand(4, B) < 0.15] = ignore_index

    def ref(p, t):
        pn = p.numpy().ravel()
        if kind == "logits":
            pn = 1 / (1 + np.exp(-pn))
        pl = (pn > 0.5).astype(int)
        tl = t.numpy().ravel()
        if ignore_index is not None:
            keep = tl != ignore_index
            pl, tl = pl[keep], tl[keep]
        return sk_fn(tl, pl)

    args = {"ignore_index": ignore_index}
    run_class_metric_test(cls, ref, preds, target, args)
    run_class_metric_ddp_test(cls, ref, preds, target, args)


@pytest.mark.parametrize("ignore_index", [None, -1])
def test_binary_auroc_ap_exact_sweep(ignore_index):
    seed_all(18)
    preds = torch.rand(4, B)
    target = torch.randint(0, 2, (4, B))
    if ignore_index is not None:
        target[torch.rand(4, B) < 0.1] = ignore_index

    def _filt(p, t):
        pn, tl = p.numpy().ravel(), t.numpy().ravel()
        if ignore_index is not None:
            keep = tl != ignore_index
            pn, tl = pn[keep], tl[keep]
        return pn, tl

    def ref_auroc(p, t):
        pn, tl = _filt(p, t)
        return skm.roc_auc_score(tl, pn)

    def ref_ap(p, t):
        pn, tl = _filt(p, t)
        return skm.average_precision_score(tl, pn)

    args = {"thresholds": None, "ignore_index": ignore_index}
    run_class_metric_test(ma.BinaryAUROC, ref_auroc, preds, target, args, check_batch=False)
    run_class_metric_ddp_test(ma.BinaryAUROC, ref_auroc, preds, target, args)
    run_class_metric_test(ma.BinaryAveragePrecision, ref_ap, preds, target, args, check_batch=False)
    run_class_metric_ddp_test(ma.BinaryAveragePrecision, ref_ap, preds, target, args)


# ---------------------------------------------------------------- multilabel
@pytest.mark.parametrize("ignore_index", [None, -1])
@pytest.mark.parametrize("average", ["micro", "macro", None])
@pytest.mark.parametrize(
    ("cls", "sk_name"),
    [
        (ma.MultilabelPrecision, "precision"),
        (ma.MultilabelRecall, "recall"),
        (ma.MultilabelF1Score, "f1"),
    ],
)
def test_multilabel_sweep(cls, sk_name, average, ignore_index):
    seed_all(19)
    preds = torch.rand(4, B, L)
    target = torch.randint(0, 2, (4, B, L))
    if ignore_index is not None:
        target[torch.rand(4, B, L) < 0.1] = ignore_index
    fns = {"precision": skm.precision_score, "recall": skm.recall_score, "f1": skm.f1_score}
    sk_fn = fns[sk_name]

    def ref(p, t):
        pl = (p.numpy() > 0.5).astype(int).reshape(-1, L)
        tl = t.numpy().reshape(-1, L)
        # per-label with ignore filtering, then average like torchmetrics
        per_label, support = [], []
        for j in range(L):
            tj, pj = tl[:, j], pl[:, j]
            if ignore_index is not None:
                keep = tj != ignore_index
                tj, pj = tj[keep], pj[keep]
            per_label.append(sk_fn(tj, pj, zero_division=0))
            support.append((tj == 1).sum())
        per_label = np.array(per_label)
        if average == "macro":
            return per_label.mean()
        if average is None:
            return per_label
        # micro: pool all labels
        if ignore_index is not None:
            keep = tl.ravel() != ignore_index
            return sk_fn(tl.ravel()[keep], pl.ravel()[keep], zero_division=0)
        return sk_fn(tl.ravel(), pl.ravel(), zero_division=0)

    args = {"num_labels": L, "average": average, "ignore_index": ignore_index}
    run_class_metric_test(cls, ref, preds, target, args)
    run_class_metric_ddp_test(cls, ref, preds, target, args)


# ---------------------------------------------------------------- regression
def _reg_inputs(seed, shape=(4, B)):
    seed_all(seed)
    preds = torch.randn(*shape)
    target = 0.7 * preds + 0.5 * torch.randn(*shape)
    return preds, target


@pytest.mark.parametrize(
    ("cls", "ref_fn", "args"),
    [
        (ma.MeanSquaredError, lambda p, t: skm.mean_squared_error(t.numpy().ravel(), p.numpy().ravel()), {}),
        (ma.MeanAbsoluteError, lambda p, t: skm.mean_absolute_error(t.numpy().ravel(), p.numpy().ravel()), {}),
        (ma.R2Score, lambda p, t: skm.r2_score(t.numpy().ravel(), p.numpy().ravel()), {}),
        (
            ma.ExplainedVariance,
            lambda p, t: skm.explained_variance_score(t.numpy().ravel(), p.numpy().ravel()),
            {},
        ),
        (
            ma.PearsonCorrCoef,
            lambda p, t: np.corrcoef(p.numpy().ravel(), t.numpy().ravel())[0, 1],
            {},
        ),
    ],
)
def test_regression_sweep(cls, ref_fn, args):
    preds, target = _reg_inputs(20)
    run_class_metric_test(cls, ref_fn, preds, target, args, check_batch=False)
    run_class_metric_ddp_test(cls, ref_fn, preds, target, args)
    run_dtype_test(cls, preds, target, args)


@pytest.mark.parametrize(
    ("cls", "sp_name"),
    [(ma.SpearmanCorrCoef, "spearmanr"), (ma.KendallRankCorrCoef, "kendalltau")],
)
def test_rank_correlation_sweep(cls, sp_name):
    from scipy import stats

    preds, target = _reg_inputs(21)
    sp_fn = getattr(stats, sp_name)

    def ref(p, t):
        return sp_fn(p.numpy().ravel(), t.numpy().ravel())[0]

    run_class_metric_test(cls, ref, preds, target, {}, check_batch=False)
    run_class_metric_ddp_test(cls, ref, preds, target, {})


# ------------------------------------------------------------ differentiable
@pytest.mark.parametrize(
    "cls",
    [ma.MeanSquaredError, ma.MeanAbsoluteError, ma.CosineSimilarity],
)
def test_forward_differentiable(cls):
    """forward() is the differentiable path: grads flow to the batch inputs."""
    seed_all(22)
    preds = torch.randn(16, 8, requires_grad=True)
    target = torch.randn(16, 8)
    m = cls()
    val = m(preds, target)
    assert val.requires_grad == bool(m.is_differentiable)
    if m.is_differentiable:
        val.sum().backward()
        assert preds.grad is not None and torch.isfinite(preds.grad).all()


# ----------------------------------------------------------------- clustering
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.clustering.MutualInfoScore, skm.mutual_info_score),
        (ma.clustering.AdjustedRandScore, skm.adjusted_rand_score),
        (ma.clustering.NormalizedMutualInfoScore, skm.normalized_mutual_info_score),
        (ma.clustering.FowlkesMallowsIndex, skm.fowlkes_mallows_score),
    ],
)
def test_clustering_sweep(cls, sk_fn):
    seed_all(23)
    preds = torch.randint(0, 6, (4, B))
    target = torch.randint(0, 5, (4, B))

    def ref(p, t):
        return sk_fn(t.numpy().ravel(), p.numpy().ravel())

    run_class_metric_test(cls, ref, preds, target, {}, check_batch=False)
    run_class_metric_ddp_test(cls, ref, preds, target, {})
