"""Aggregation metric tests."""
import pytest
import torch

from metrics_amd import CatMetric, MaxMetric, MeanMetric, MinMetric, RunningMean, RunningSum, SumMetric


def test_sum_metric():
    m = SumMetric()
    m.update(1.0)
    m.update(torch.tensor([2.0, 3.0]))
    assert m.compute() == 6.0


def test_max_min_metric():
    mx, mn = MaxMetric(), MinMetric()
    for v in (1.0, 5.0, 3.0):
        mx.update(v)
        mn.update(v)
    assert mx.compute() == 5.0
    assert mn.compute() == 1.0


def test_cat_metric():
    m = CatMetric()
    m.update(torch.tensor([1.0, 2.0]))
    m.update(3.0)
    assert torch.equal(m.compute(), torch.tensor([1.0, 2.0, 3.0]))


def test_mean_metric_weighted():
    m = MeanMetric()
    m.update(2.0, weight=1.0)
    m.update(4.0, weight=3.0)
    assert m.compute() == (2 + 12) / 4


def test_nan_strategies():
    with pytest.raises(RuntimeError, match="nan"):
        m = SumMetric(nan_strategy="error")
        m.update(torch.tensor([1.0, float("nan")]))
    m = SumMetric(nan_strategy="ignore")
    m.update(torch.tensor([1.0, float("nan"), 2.0]))
    assert m.compute() == 3.0
    m = SumMetric(nan_strategy=0.0)
    m.update(torch.tensor([1.0, float("nan")]))
    assert m.compute() == 1.0
    with pytest.raises(ValueError, match="nan_strategy"):
        SumMetric(nan_strategy="bad")


def test_running_mean():
    m = RunningMean(window=3)
    vals = [1.0, 2.0, 3.0, 4.0, 5.0]
    outs = []
    for v in vals:
        m.update(v)
        outs.append(m.compute().item())
    assert outs[0] == 1.0
    assert outs[2] == pytest.approx(2.0)
    assert outs[4] == pytest.approx(4.0)  # mean of 3,4,5


def test_running_sum():
    m = RunningSum(window=2)
    for v in (1.0, 2.0, 3.0):
        m.update(v)
    assert m.compute() == 5.0


def test_forward_aggregation():
    m = MeanMetric()
    out = m(torch.tensor([2.0, 4.0]))
    assert out == 3.0
    m(torch.tensor([6.0]))
    assert m.compute() == 4.0
