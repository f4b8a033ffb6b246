"""sklearn-oracle tests for curve metrics (ROC / PRC / AUROC / AP / calibration)."""
import numpy as np
import pytest
import torch
from sklearn import metrics as skm

import metrics_amd as ma
from tests.unittests._helpers import seed_all

NUM_CLASSES = 5


@pytest.fixture()
def bin_scores():
    seed_all(11)
    return torch.rand(512), torch.randint(0, 2, (512,))


@pytest.fixture()
def mc_scores():
    seed_all(12)
    return torch.randn(512, NUM_CLASSES).softmax(-1), torch.randint(0, NUM_CLASSES, (512,))


def test_binary_roc_exact(bin_scores):
    preds, target = bin_scores
    fpr, tpr, thr = ma.BinaryROC()(preds, target)
    s_fpr, s_tpr, s_thr = skm.roc_curve(target.numpy(), preds.numpy(), drop_intermediate=False)
    assert np.allclose(fpr.numpy(), s_fpr)
    assert np.allclose(tpr.numpy(), s_tpr)


def test_binary_auroc_exact_and_thresholded(bin_scores):
    preds, target = bin_scores
    ref = skm.roc_auc_score(target.numpy(), preds.numpy())
    assert abs(ma.BinaryAUROC()(preds, target).item() - ref) < 1e-6
    assert abs(ma.BinaryAUROC(thresholds=5000)(preds, target).item() - ref) < 2e-3


def test_binary_average_precision(bin_scores):
    preds, target = bin_scores
    ref = skm.average_precision_score(target.numpy(), preds.numpy())
    assert abs(ma.BinaryAveragePrecision()(preds, target).item() - ref) < 1e-6


def test_binary_prc_exact(bin_scores):
    preds, target = bin_scores
    p, r, t = ma.BinaryPrecisionRecallCurve()(preds, target)
    sp, sr, st = skm.precision_recall_curve(target.numpy(), preds.numpy())
    assert np.allclose(p.numpy(), sp, atol=1e-6)
    assert np.allclose(r.numpy(), sr, atol=1e-6)


def test_multiclass_auroc(mc_scores):
    preds, target = mc_scores
    for avg in ("macro", "weighted"):
        ref = skm.roc_auc_score(target.numpy(), preds.numpy(), multi_class="ovr", average=avg, labels=range(NUM_CLASSES))
        v = ma.MulticlassAUROC(num_classes=NUM_CLASSES, average=avg)(preds, target).item()
        assert abs(v - ref) < 1e-5, (avg, v, ref)


def test_multiclass_auroc_thresholded_close(mc_scores):
    preds, target = mc_scores
    exact = ma.MulticlassAUROC(num_classes=NUM_CLASSES, average="macro")(preds, target).item()
    approx = ma.MulticlassAUROC(num_classes=NUM_CLASSES, average="macro", thresholds=5000)(preds, target).item()
    assert abs(exact - approx) < 2e-3


def test_multiclass_average_precision(mc_scores):
    preds, target = mc_scores
    onehot = torch.nn.functional.one_hot(target, NUM_CLASSES).numpy()
    ref = skm.average_precision_score(onehot, preds.numpy(), average="macro")
    v = ma.MulticlassAveragePrecision(num_classes=NUM_CLASSES, average="macro")(preds, target).item()
    assert abs(v - ref) < 1e-5


def test_thresholded_state_matches_batch_accumulation(bin_scores):
    """Accumulating the (T,2,2) confmat over batches == single-shot update."""
    preds, target = bin_scores
    m1 = ma.BinaryPrecisionRecallCurve(thresholds=100)
    m2 = ma.BinaryPrecisionRecallCurve(thresholds=100)
    m1.update(preds, target)
    for chunk in range(4):
        m2.update(preds[chunk * 128:(chunk + 1) * 128], target[chunk * 128:(chunk + 1) * 128])
    assert torch.equal(m1.confmat, m2.confmat)


def test_binary_calibration_error(bin_scores):
    preds, target = bin_scores
    v = ma.BinaryCalibrationError(n_bins=10, norm="l1")(preds, target).item()
    # manual ECE oracle (reference semantics: confidence = p(class 1),
    # accuracy = target; see reference functional/classification/calibration_error.py)
    conf = preds.float()
    acc = target.float()
    bins = torch.linspace(0, 1, 11)
    idx = (torch.bucketize(conf, bins, right=True) - 1).clamp(0, 10)
    ece = 0.0
    for b in range(11):
        mask = idx == b
        if mask.any():
            ece += (mask.float().mean() * (acc[mask].mean() - conf[mask].mean()).abs()).item()
    assert abs(v - ece) < 1e-6


def test_multilabel_auroc():
    seed_all(13)
    preds = torch.rand(256, 4)
    target = torch.randint(0, 2, (256, 4))
    ref = skm.roc_auc_score(target.numpy(), preds.numpy(), average="macro")
    v = ma.MultilabelAUROC(num_labels=4, average="macro")(preds, target).item()
    assert abs(v - ref) < 1e-5


def test_multilabel_ranking():
    seed_all(14)
    preds = torch.rand(64, 4)
    target = torch.randint(0, 2, (64, 4))
    v = ma.MultilabelCoverageError(num_labels=4)(preds, target).item()
    ref = skm.coverage_error(target.numpy(), preds.numpy())
    assert abs(v - ref) < 1e-5
    v = ma.MultilabelRankingAveragePrecision(num_labels=4)(preds, target).item()
    ref = skm.label_ranking_average_precision_score(target.numpy(), preds.numpy())
    assert abs(v - ref) < 1e-5
    v = ma.MultilabelRankingLoss(num_labels=4)(preds, target).item()
    ref = skm.label_ranking_loss(target.numpy(), preds.numpy())
    assert abs(v - ref) < 1e-5


def test_hinge_loss():
    seed_all(15)
    preds = torch.randn(128)
    target = torch.randint(0, 2, (128,))
    # logits are auto-sigmoided by the format stage (reference semantics),
    # so the sklearn oracle gets the sigmoided scores
    v = ma.BinaryHingeLoss()(preds, target).item()
    ref = skm.hinge_loss(target.numpy(), torch.sigmoid(preds).numpy())
    assert abs(v - ref) < 1e-5
