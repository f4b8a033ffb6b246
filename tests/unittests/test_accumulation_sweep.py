"""Broad invariance sweep: for many metrics, (a) updating in two chunks equals
one full update (state additivity), and (b) the modular result equals the
functional twin on the full data. Mirrors the reference's modular/functional
agreement strategy without porting its fixtures."""
import pytest
import torch

import metrics_amd as ma
import metrics_amd.functional as F

from tests.unittests._helpers import seed_all

B, C, L = 199, 7, 5


def _mc():
    seed_all(41)
    return torch.randn(B, C).softmax(1), torch.randint(0, C, (B,))


def _bin():
    seed_all(42)
    return torch.rand(B), torch.randint(0, 2, (B,))


def _ml():
    seed_all(43)
    return torch.rand(B, L), torch.randint(0, 2, (B, L))


def _reg():
    seed_all(44)
    return torch.randn(B), torch.randn(B)


CASES = [
    # (modular ctor, functional fn(preds, target), input builder)
    (lambda: ma.MulticlassAccuracy(num_classes=C, average="macro"),
     lambda p, t: F.accuracy(p, t, task="multiclass", num_classes=C, average="macro"), _mc),
    (lambda: ma.MulticlassPrecision(num_classes=C, average="weighted"),
     lambda p, t: F.precision(p, t, task="multiclass", num_classes=C, average="weighted"), _mc),
    (lambda: ma.MulticlassRecall(num_classes=C, average="micro"),
     lambda p, t: F.recall(p, t, task="multiclass", num_classes=C, average="micro"), _mc),
    (lambda: ma.MulticlassF1Score(num_classes=C, average="none"),
     lambda p, t: F.f1_score(p, t, task="multiclass", num_classes=C, average="none"), _mc),
    (lambda: ma.MulticlassSpecificity(num_classes=C, average="macro"),
     lambda p, t: F.specificity(p, t, task="multiclass", num_classes=C, average="macro"), _mc),
    (lambda: ma.MulticlassConfusionMatrix(num_classes=C),
     lambda p, t: F.confusion_matrix(p, t, task="multiclass", num_classes=C), _mc),
    (lambda: ma.MulticlassCohenKappa(num_classes=C),
     lambda p, t: F.cohen_kappa(p, t, task="multiclass", num_classes=C), _mc),
    (lambda: ma.MulticlassMatthewsCorrCoef(num_classes=C),
     lambda p, t: F.matthews_corrcoef(p, t, task="multiclass", num_classes=C), _mc),
    (lambda: ma.MulticlassJaccardIndex(num_classes=C),
     lambda p, t: F.jaccard_index(p, t, task="multiclass", num_classes=C), _mc),
    (lambda: ma.MulticlassAUROC(num_classes=C, thresholds=50),
     lambda p, t: F.auroc(p, t, task="multiclass", num_classes=C, thresholds=50), _mc),
    (lambda: ma.MulticlassAveragePrecision(num_classes=C, thresholds=50),
     lambda p, t: F.average_precision(p, t, task="multiclass", num_classes=C, thresholds=50), _mc),
    (lambda: ma.MulticlassCalibrationError(num_classes=C, n_bins=10),
     lambda p, t: F.calibration_error(p, t, task="multiclass", num_classes=C, n_bins=10), _mc),
    (lambda: ma.MulticlassHingeLoss(num_classes=C),
     lambda p, t: F.hinge_loss(p, t, task="multiclass", num_classes=C), _mc),
    (lambda: ma.BinaryAccuracy(), lambda p, t: F.accuracy(p, t, task="binary"), _bin),
    (lambda: ma.BinaryAUROC(), lambda p, t: F.auroc(p, t, task="binary"), _bin),
    (lambda: ma.BinaryCalibrationError(n_bins=12),
     lambda p, t: F.calibration_error(p, t, task="binary", n_bins=12), _bin),
    (lambda: ma.MultilabelF1Score(num_labels=L, average="macro"),
     lambda p, t: F.f1_score(p, t, task="multilabel", num_labels=L, average="macro"), _ml),
    (lambda: ma.MultilabelRankingAveragePrecision(num_labels=L),
     lambda p, t: F.multilabel_ranking_average_precision(p, t, num_labels=L), _ml),
    (lambda: ma.MeanSquaredError(), F.mean_squared_error, _reg),
    (lambda: ma.MeanAbsoluteError(), F.mean_absolute_error, _reg),
    (lambda: ma.PearsonCorrCoef(), F.pearson_corrcoef, _reg),
    (lambda: ma.SpearmanCorrCoef(), F.spearman_corrcoef, _reg),
    (lambda: ma.R2Score(), F.r2_score, _reg),
    (lambda: ma.ExplainedVariance(), F.explained_variance, _reg),
    (lambda: ma.ConcordanceCorrCoef(), F.concordance_corrcoef, _reg),
    (lambda: ma.KendallRankCorrCoef(), F.kendall_rank_corrcoef, _reg),
    (lambda: ma.LogCoshError(), F.log_cosh_error, _reg),
    (lambda: ma.CosineSimilarity(reduction="mean"),
     lambda p, t: F.cosine_similarity(p.unsqueeze(0), t.unsqueeze(0), reduction="mean"), _reg),
]


def _flatcmp(a, b, atol=1e-5):
    if isinstance(a, (tuple, list)):
        assert len(a) == len(b)
        for x, y in zip(a, b):
            _flatcmp(x, y, atol)
    else:
        assert torch.allclose(a, b, atol=atol), (a, b)


@pytest.mark.parametrize("case", range(len(CASES)))
def test_chunked_equals_full_and_functional(case):
    mod_ctor, fn, data = CASES[case]
    preds, target = data()
    if mod_ctor().__class__.__name__ == "CosineSimilarity":
        preds, target = preds.unsqueeze(0), target.unsqueeze(0)

    full = mod_ctor()
    full.update(preds, target)
    r_full = full.compute()

    chunked = mod_ctor()
    cut = B // 3
    chunked.update(preds[:cut], target[:cut])
    chunked.update(preds[cut:], target[cut:])
    r_chunked = chunked.compute()
    _flatcmp(r_full, r_chunked, atol=1e-4)

    if mod_ctor().__class__.__name__ == "CosineSimilarity":
        r_fn = fn(preds.squeeze(0), target.squeeze(0))
    else:
        r_fn = fn(preds, target)
    _flatcmp(r_full, r_fn, atol=1e-4)
