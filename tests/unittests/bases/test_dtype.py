"""Precision tests: metrics must work in double (CPU) and keep results consistent.

Mirrors the reference's ``run_precision_test_cpu`` strategy (half is covered on
GPU in tests/unittests/gpu/).
"""
import pytest
import torch

import metrics_amd as ma
from tests.unittests._helpers import seed_all


CLS_CASES = [
    lambda: ma.MulticlassAccuracy(num_classes=5, average="macro"),
    lambda: ma.MulticlassF1Score(num_classes=5, average="weighted"),
    lambda: ma.BinaryAUROC(thresholds=20),
    lambda: ma.MulticlassCalibrationError(num_classes=5),
    lambda: ma.MulticlassCohenKappa(num_classes=5),
]


@pytest.mark.parametrize("make", CLS_CASES)
def test_classification_double(make):
    seed_all(50)
    preds32 = torch.randn(128, 5).softmax(-1)
    target = torch.randint(0, 5, (128,))
    p = preds32 if not isinstance(make(), (ma.BinaryAUROC,)) else None
    m32, m64 = make(), make().set_dtype(torch.double)
    if isinstance(m32, ma.BinaryAUROC):
        preds = torch.rand(128)
        m32.update(preds, target.clamp(max=1))
        m64.update(preds.double(), target.clamp(max=1))
    else:
        m32.update(preds32, target)
        m64.update(preds32.double(), target)
    r32, r64 = m32.compute(), m64.compute()
    assert r64.dtype in (torch.float64, torch.long) or r64.dtype == r32.dtype
    assert torch.allclose(r32.double(), r64.double(), atol=1e-5)


REG_CASES = [
    lambda: ma.MeanSquaredError(),
    lambda: ma.MeanAbsoluteError(),
    lambda: ma.PearsonCorrCoef(),
    lambda: ma.SpearmanCorrCoef(),
    lambda: ma.R2Score(),
    lambda: ma.ExplainedVariance(),
]


@pytest.mark.parametrize("make", REG_CASES)
def test_regression_double(make):
    seed_all(51)
    x = torch.randn(200)
    y = 0.7 * x + 0.2 * torch.randn(200)
    m32, m64 = make(), make().set_dtype(torch.double)
    m32.update(x, y)
    m64.update(x.double(), y.double())
    assert torch.allclose(m32.compute().double(), m64.compute().double(), atol=1e-4)


def test_half_state_conversion_roundtrip():
    m = ma.MeanSquaredError().set_dtype(torch.half)
    assert m.sum_squared_error.dtype == torch.half
    m = m.set_dtype(torch.float32)
    assert m.sum_squared_error.dtype == torch.float32


def test_double_preserved_through_reset():
    m = ma.MeanSquaredError().set_dtype(torch.double)
    m.update(torch.randn(8).double(), torch.randn(8).double())
    m.reset()
    assert m.sum_squared_error.dtype == torch.float64


def test_image_metrics_double():
    seed_all(52)
    p = torch.rand(1, 3, 32, 32)
    t = torch.rand(1, 3, 32, 32)
    m32 = ma.StructuralSimilarityIndexMeasure(data_range=1.0)
    m64 = ma.StructuralSimilarityIndexMeasure(data_range=1.0).set_dtype(torch.double)
    assert torch.allclose(m32(p, t).double(), m64(p.double(), t.double()), atol=1e-4)
    p32 = ma.PeakSignalNoiseRatio(data_range=1.0)
    p64 = ma.PeakSignalNoiseRatio(data_range=1.0).set_dtype(torch.double)
    assert torch.allclose(p32(p, t).double(), p64(p.double(), t.double()), atol=1e-4)
