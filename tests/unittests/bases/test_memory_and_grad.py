"""Constant-memory update checks + gradcheck differentiability (reference
test strategy: constant-memory tests in bases/test_metric.py:433-496,
run_differentiability_test in _helpers/testers.py:553-588)."""
import pytest
import torch

import metrics_amd as ma
from tests.unittests._helpers import seed_all


def _state_bytes(metric):
    total = 0
    for name in metric._defaults:
        v = getattr(metric, name)
        if isinstance(v, torch.Tensor):
            total += v.numel() * v.element_size()
        else:
            total += sum(t.numel() * t.element_size() for t in v)
    return total


@pytest.mark.parametrize(
    "make",
    [
        lambda: ma.MulticlassAccuracy(num_classes=10),
        lambda: ma.MulticlassConfusionMatrix(num_classes=10),
        lambda: ma.BinaryAUROC(thresholds=32),
        lambda: ma.MeanSquaredError(),
        lambda: ma.PearsonCorrCoef(),
        lambda: ma.MeanMetric(),
    ],
)
def test_constant_memory_fixed_state_metrics(make):
    """Fixed-shape-state metrics must not grow state with more updates."""
    seed_all(91)
    m = make()

    def feed():
        if isinstance(m, (ma.MeanSquaredError, ma.PearsonCorrCoef)):
            m.update(torch.randn(32), torch.randn(32))
        elif isinstance(m, ma.MeanMetric):
            m.update(torch.randn(32))
        elif isinstance(m, ma.BinaryAUROC):
            m.update(torch.rand(32), torch.randint(0, 2, (32,)))
        else:
            m.update(torch.randn(32, 10), torch.randint(0, 10, (32,)))

    feed()
    base = _state_bytes(m)
    for _ in range(50):
        feed()
    assert _state_bytes(m) == base


def test_list_state_metrics_grow_linearly():
    m = ma.SpearmanCorrCoef()
    m.update(torch.randn(32), torch.randn(32))
    b1 = _state_bytes(m)
    m.update(torch.randn(32), torch.randn(32))
    assert _state_bytes(m) == 2 * b1


@pytest.mark.parametrize(
    ("fn", "n_args"),
    [
        (lambda p, t: __import__("metrics_amd").functional.mean_squared_error(p, t), 2),
        (lambda p, t: __import__("metrics_amd").functional.mean_absolute_error(p, t), 2),
        (lambda p, t: __import__("metrics_amd").functional.explained_variance(p, t), 2),
    ],
)
def test_gradcheck_differentiable_functionals(fn, n_args):
    seed_all(92)
    p = torch.randn(8, 4, dtype=torch.double, requires_grad=True)
    t = torch.randn(8, 4, dtype=torch.double)
    assert torch.autograd.gradcheck(lambda x: fn(x, t), (p,), fast_mode=True)


def test_metric_forward_differentiable():
    m = ma.MeanSquaredError()
    p = torch.randn(16, requires_grad=True)
    t = torch.randn(16)
    out = m(p, t)
    assert out.requires_grad
    out.backward()
    assert p.grad is not None
    # state must be detached
    assert not m.sum_squared_error.requires_grad
