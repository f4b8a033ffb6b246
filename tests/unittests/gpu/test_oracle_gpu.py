"""GPU-direct sklearn-oracle tests.

The CPU oracle sweep validates our CPU path against sklearn; the GPU kernel
tests validate HIP against our CPU path. This module closes the remaining
link directly — HIP kernels vs sklearn on the SAME inputs — so a shared bug
in the (common) format/compute stages cannot hide.
"""
import numpy as np
import pytest
import torch
from sklearn import metrics as skm

pytestmark = pytest.mark.gpu

import metrics_amd as ma

C = 9
B = 4096


def _mc_data(seed=50):
    g = torch.Generator().manual_seed(seed)
    preds = torch.randn(B, C, generator=g)
    target = torch.randint(0, C, (B,), generator=g)
    return preds, target


@pytest.mark.parametrize("average", ["micro", "macro", "weighted"])
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.MulticlassPrecision, skm.precision_score),
        (ma.MulticlassRecall, skm.recall_score),
        (ma.MulticlassF1Score, skm.f1_score),
    ],
)
def test_gpu_mc_prf_vs_sklearn(cls, sk_fn, average):
    preds, target = _mc_data()
    m = cls(num_classes=C, average=average).to("cuda")
    m.update(preds.cuda(), target.cuda())
    got = m.compute().cpu()
    exp = sk_fn(target.numpy(), preds.argmax(-1).numpy(), labels=range(C), average=average, zero_division=0)
    assert abs(float(got) - exp) < 1e-5


@pytest.mark.parametrize("ignore_index", [None, 3])
def test_gpu_mc_accuracy_confmat_vs_sklearn(ignore_index):
    preds, target = _mc_data(51)
    if ignore_index is not None:
        target[::11] = ignore_index
    m = ma.MulticlassAccuracy(num_classes=C, average="micro", ignore_index=ignore_index).to("cuda")
    m.update(preds.cuda(), target.cuda())
    cm = ma.MulticlassConfusionMatrix(num_classes=C, ignore_index=ignore_index).to("cuda")
    cm.update(preds.cuda(), target.cuda())
    tl, pl = target.numpy(), preds.argmax(-1).numpy()
    if ignore_index is not None:
        keep = tl != ignore_index
        tl, pl = tl[keep], pl[keep]
    assert abs(float(m.compute()) - skm.accuracy_score(tl, pl)) < 1e-6
    assert np.array_equal(cm.compute().cpu().numpy(), skm.confusion_matrix(tl, pl, labels=range(C)))


def test_gpu_binary_vs_sklearn():
    g = torch.Generator().manual_seed(52)
    preds = torch.rand(B, generator=g)
    target = torch.randint(0, 2, (B,), generator=g)
    pl = (preds.numpy() > 0.5).astype(int)
    for cls, sk in [
        (ma.BinaryAccuracy, lambda t, p: skm.accuracy_score(t, p)),
        (ma.BinaryF1Score, lambda t, p: skm.f1_score(t, p, zero_division=0)),
        (ma.BinaryMatthewsCorrCoef, skm.matthews_corrcoef),
    ]:
        m = cls().to("cuda")
        m.update(preds.cuda(), target.cuda())
        assert abs(float(m.compute()) - sk(target.numpy(), pl)) < 1e-5, cls.__name__


def test_gpu_auroc_ap_vs_sklearn_exact_and_bucketized():
    g = torch.Generator().manual_seed(53)
    preds = torch.rand(B, generator=g)
    target = torch.randint(0, 2, (B,), generator=g)
    sk_auroc = skm.roc_auc_score(target.numpy(), preds.numpy())
    sk_ap = skm.average_precision_score(target.numpy(), preds.numpy())
    # exact (K2 sort kernel)
    m = ma.BinaryAUROC(thresholds=None).to("cuda")
    m.update(preds.cuda(), target.cuda())
    assert abs(float(m.compute()) - sk_auroc) < 1e-5
    a = ma.BinaryAveragePrecision(thresholds=None).to("cuda")
    a.update(preds.cuda(), target.cuda())
    assert abs(float(a.compute()) - sk_ap) < 1e-5
    # bucketized (K5-family histogram kernel) — binning error bounded
    mb = ma.BinaryAUROC(thresholds=2000).to("cuda")
    mb.update(preds.cuda(), target.cuda())
    assert abs(float(mb.compute()) - sk_auroc) < 2e-3


def test_gpu_multiclass_auroc_vs_sklearn():
    preds, target = _mc_data(54)
    probs = preds.softmax(-1)
    m = ma.MulticlassAUROC(num_classes=C, average="macro", thresholds=None).to("cuda")
    m.update(probs.cuda(), target.cuda())
    exp = skm.roc_auc_score(target.numpy(), probs.numpy(), multi_class="ovr", average="macro", labels=range(C))
    assert abs(float(m.compute()) - exp) < 1e-5


def test_gpu_regression_vs_sklearn():
    g = torch.Generator().manual_seed(55)
    preds = torch.randn(B, generator=g)
    target = 0.7 * preds + 0.5 * torch.randn(B, generator=g)
    for cls, sk in [
        (ma.MeanSquaredError, skm.mean_squared_error),
        (ma.MeanAbsoluteError, skm.mean_absolute_error),
        (ma.R2Score, skm.r2_score),
    ]:
        m = cls().to("cuda")
        m.update(preds.cuda(), target.cuda())
        assert abs(float(m.compute()) - sk(target.numpy(), preds.numpy())) < 1e-5, cls.__name__
    p = ma.PearsonCorrCoef().to("cuda")
    p.update(preds.cuda(), target.cuda())
    assert abs(float(p.compute()) - np.corrcoef(preds.numpy(), target.numpy())[0, 1]) < 1e-5


def test_gpu_topk_vs_sklearn():
    preds, target = _mc_data(56)
    probs = preds.softmax(-1)
    m = ma.MulticlassAccuracy(num_classes=C, average="micro", top_k=3).to("cuda")
    m.update(probs.cuda(), target.cuda())
    exp = skm.top_k_accuracy_score(target.numpy(), probs.numpy(), k=3, labels=range(C))
    assert abs(float(m.compute()) - exp) < 1e-6


def test_gpu_calibration_vs_manual_numpy():
    g = torch.Generator().manual_seed(57)
    conf = torch.rand(B, generator=g)
    target = torch.randint(0, 2, (B,), generator=g)
    m = ma.BinaryCalibrationError(n_bins=15, norm="l1").to("cuda")
    m.update(conf.cuda(), target.cuda())
    # independent numpy ECE (reference convention: confidences are the raw
    # class-1 probabilities, accuracies the 0/1 targets)
    c = conf.numpy()
    acc = target.numpy().astype(float)
    edges = np.linspace(0, 1, 16)
    idx = np.clip(np.searchsorted(edges, c, side="right") - 1, 0, 14)
    ece = 0.0
    for b in range(15):
        sel = idx == b
        if sel.any():
            ece += abs(acc[sel].mean() - c[sel].mean()) * sel.mean()
    assert abs(float(m.compute()) - ece) < 1e-4
