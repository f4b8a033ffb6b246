"""sklearn-oracle tests for the stat-score metric family (CPU torch path).

Strategy mirrors the reference MetricTester: seeded random inputs, batch-wise
forward + accumulated compute compared against sklearn on all data.
"""
import numpy as np
import pytest
import torch
from sklearn import metrics as skm

import metrics_amd as ma
from tests.unittests._helpers import run_class_metric_test, seed_all

NUM_CLASSES = 5
NUM_LABELS = 4
B = 33  # deliberately not a multiple of anything


@pytest.fixture()
def mc_data():
    seed_all(3)
    preds = torch.randn(4, B, NUM_CLASSES)
    target = torch.randint(0, NUM_CLASSES, (4, B))
    return preds, target


@pytest.fixture()
def bin_data():
    seed_all(4)
    preds = torch.rand(4, B)
    target = torch.randint(0, 2, (4, B))
    return preds, target


@pytest.fixture()
def ml_data():
    seed_all(5)
    preds = torch.rand(4, B, NUM_LABELS)
    target = torch.randint(0, 2, (4, B, NUM_LABELS))
    return preds, target


@pytest.mark.parametrize("average", ["micro", "macro", "weighted", None])
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.MulticlassPrecision, skm.precision_score),
        (ma.MulticlassRecall, skm.recall_score),
        (ma.MulticlassF1Score, skm.f1_score),
    ],
)
def test_multiclass_prf(mc_data, cls, sk_fn, average):
    preds, target = mc_data

    def ref(p, t):
        return sk_fn(
            t.numpy(), p.argmax(-1).numpy(), labels=range(NUM_CLASSES),
            average=average, zero_division=0,
        )

    run_class_metric_test(cls, ref, preds, target, {"num_classes": NUM_CLASSES, "average": average})


@pytest.mark.parametrize("average", ["micro", "macro", "weighted"])
def test_multiclass_accuracy(mc_data, average):
    preds, target = mc_data

    def ref(p, t):
        if average == "micro":
            return skm.accuracy_score(t.numpy(), p.argmax(-1).numpy())
        return skm.recall_score(
            t.numpy(), p.argmax(-1).numpy(), labels=range(NUM_CLASSES), average=average, zero_division=0
        )

    run_class_metric_test(ma.MulticlassAccuracy, ref, preds, target, {"num_classes": NUM_CLASSES, "average": average})


def test_multiclass_confmat(mc_data):
    preds, target = mc_data

    def ref(p, t):
        return skm.confusion_matrix(t.numpy(), p.argmax(-1).numpy(), labels=range(NUM_CLASSES))

    run_class_metric_test(ma.MulticlassConfusionMatrix, ref, preds, target, {"num_classes": NUM_CLASSES})


def test_multiclass_ignore_index():
    seed_all(6)
    preds = torch.randn(128, NUM_CLASSES)
    target = torch.randint(0, NUM_CLASSES, (128,))
    target[::7] = -1
    m = ma.MulticlassAccuracy(num_classes=NUM_CLASSES, average="micro", ignore_index=-1)
    v = m(preds, target)
    keep = target != -1
    ref = skm.accuracy_score(target[keep].numpy(), preds.argmax(-1)[keep].numpy())
    assert abs(v.item() - ref) < 1e-6


def test_multiclass_top_k():
    seed_all(7)
    preds = torch.randn(256, NUM_CLASSES).softmax(-1)
    target = torch.randint(0, NUM_CLASSES, (256,))
    v = ma.MulticlassAccuracy(num_classes=NUM_CLASSES, average="micro", top_k=2)(preds, target)
    ref = skm.top_k_accuracy_score(target.numpy(), preds.numpy(), k=2, labels=range(NUM_CLASSES))
    assert abs(v.item() - ref) < 1e-6, (v.item(), ref)


@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.BinaryAccuracy, skm.accuracy_score),
        (ma.BinaryPrecision, skm.precision_score),
        (ma.BinaryRecall, skm.recall_score),
        (ma.BinaryF1Score, skm.f1_score),
        (ma.BinaryMatthewsCorrCoef, skm.matthews_corrcoef),
        (ma.BinaryCohenKappa, skm.cohen_kappa_score),
        (ma.BinaryJaccardIndex, skm.jaccard_score),
    ],
)
def test_binary_metrics(bin_data, cls, sk_fn):
    preds, target = bin_data

    def ref(p, t):
        return sk_fn(t.numpy(), (p.numpy() > 0.5).astype(int))

    run_class_metric_test(cls, ref, preds, target, {})


def test_binary_specificity(bin_data):
    preds, target = bin_data

    def ref(p, t):
        tn, fp, fn, tp = skm.confusion_matrix(t.numpy(), (p.numpy() > 0.5).astype(int), labels=[0, 1]).ravel()
        return tn / (tn + fp)

    run_class_metric_test(ma.BinarySpecificity, ref, preds, target, {})


def test_binary_npv(bin_data):
    preds, target = bin_data

    def ref(p, t):
        tn, fp, fn, tp = skm.confusion_matrix(t.numpy(), (p.numpy() > 0.5).astype(int), labels=[0, 1]).ravel()
        return tn / (tn + fn)

    run_class_metric_test(ma.BinaryNegativePredictiveValue, ref, preds, target, {})


def test_binary_hamming(bin_data):
    preds, target = bin_data

    def ref(p, t):
        return 1 - skm.accuracy_score(t.numpy(), (p.numpy() > 0.5).astype(int))

    run_class_metric_test(ma.BinaryHammingDistance, ref, preds, target, {})


def test_binary_with_logits():
    """Logit inputs outside [0,1] are auto-sigmoided before thresholding."""
    seed_all(8)
    logits = torch.randn(200) * 3
    target = torch.randint(0, 2, (200,))
    v = ma.BinaryAccuracy()(logits, target)
    ref = skm.accuracy_score(target.numpy(), (torch.sigmoid(logits) > 0.5).numpy())
    assert abs(v.item() - ref) < 1e-6


@pytest.mark.parametrize("average", ["micro", "macro", "weighted", None])
@pytest.mark.parametrize(
    ("cls", "sk_fn"),
    [
        (ma.MultilabelPrecision, skm.precision_score),
        (ma.MultilabelRecall, skm.recall_score),
        (ma.MultilabelF1Score, skm.f1_score),
    ],
)
def test_multilabel_prf(ml_data, cls, sk_fn, average):
    preds, target = ml_data

    def ref(p, t):
        return sk_fn(t.numpy(), (p.numpy() > 0.5).astype(int), average=average, zero_division=0)

    run_class_metric_test(cls, ref, preds, target, {"num_labels": NUM_LABELS, "average": average})


def test_multilabel_exact_match(ml_data):
    preds, target = ml_data

    def ref(p, t):
        return skm.accuracy_score(t.reshape(-1, NUM_LABELS).numpy(), (p.reshape(-1, NUM_LABELS).numpy() > 0.5).astype(int))

    run_class_metric_test(ma.MultilabelExactMatch, ref, preds, target, {"num_labels": NUM_LABELS})


def test_multiclass_exact_match():
    seed_all(9)
    preds = torch.randint(0, NUM_CLASSES, (4, B, 7))
    target = torch.randint(0, NUM_CLASSES, (4, B, 7))

    def ref(p, t):
        return (p.numpy() == t.numpy()).all(-1).mean()

    run_class_metric_test(ma.MulticlassExactMatch, ref, preds, target, {"num_classes": NUM_CLASSES})


def test_stat_scores_output_shape(mc_data):
    preds, target = mc_data
    m = ma.MulticlassStatScores(num_classes=NUM_CLASSES, average=None)
    m.update(preds[0], target[0])
    out = m.compute()
    assert out.shape == (NUM_CLASSES, 5)
    assert (out[:, 4] == out[:, 0] + out[:, 3]).all()  # support = tp + fn


def test_samplewise_multidim():
    seed_all(10)
    preds = torch.randint(0, NUM_CLASSES, (8, 16))
    target = torch.randint(0, NUM_CLASSES, (8, 16))
    m = ma.MulticlassAccuracy(num_classes=NUM_CLASSES, average="micro", multidim_average="samplewise")
    v = m(preds, target)
    ref = (preds == target).float().mean(dim=1)
    assert torch.allclose(v, ref)
