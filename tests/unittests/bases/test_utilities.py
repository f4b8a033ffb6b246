"""Utility-layer tests (reference tests/unittests/utilities parity)."""
import numpy as np
import pytest
import torch

from metrics_amd.utilities.checks import _check_same_shape
from metrics_amd.utilities.compute import _auc_compute_without_check, _safe_divide, _safe_matmul, _safe_xlogy, interp
from metrics_amd.utilities.data import (
    _bincount,
    _cumsum,
    _flexible_bincount,
    dim_zero_cat,
    dim_zero_max,
    dim_zero_mean,
    dim_zero_min,
    dim_zero_sum,
    select_topk,
    to_onehot,
)
from metrics_amd.utilities.distributed import class_reduce, reduce


def test_dim_zero_helpers():
    xs = [torch.tensor([1.0, 2.0]), torch.tensor([3.0, 4.0])]
    assert torch.equal(dim_zero_cat(xs), torch.tensor([1.0, 2.0, 3.0, 4.0]))
    stacked = torch.stack(xs)
    assert torch.equal(dim_zero_sum(stacked), torch.tensor([4.0, 6.0]))
    assert torch.equal(dim_zero_mean(stacked), torch.tensor([2.0, 3.0]))
    assert torch.equal(dim_zero_max(stacked), torch.tensor([3.0, 4.0]))
    assert torch.equal(dim_zero_min(stacked), torch.tensor([1.0, 2.0]))
    # single tensor passes through cat
    assert torch.equal(dim_zero_cat(torch.tensor([5.0])), torch.tensor([5.0]))


def test_to_onehot_roundtrip():
    labels = torch.tensor([0, 2, 1])
    oh = to_onehot(labels, num_classes=3)
    assert oh.shape == (3, 3)
    assert torch.equal(oh.argmax(1), labels)
    # float inputs keep dtype
    probs = torch.randn(4, 3)
    oh2 = to_onehot(probs.argmax(1), num_classes=3)
    assert oh2.sum() == 4


def test_select_topk():
    probs = torch.tensor([[0.1, 0.7, 0.2], [0.5, 0.4, 0.1]])
    t1 = select_topk(probs, topk=1)
    assert torch.equal(t1, torch.tensor([[0, 1, 0], [1, 0, 0]], dtype=t1.dtype))
    t2 = select_topk(probs, topk=2)
    assert torch.equal(t2.sum(1), torch.tensor([2, 2]))


def test_bincount_variants():
    x = torch.tensor([0, 1, 1, 2, 2, 2])
    assert torch.equal(_bincount(x, minlength=4), torch.tensor([1, 2, 3, 0]))
    assert torch.equal(_flexible_bincount(x), torch.tensor([1, 2, 3]))
    # deterministic mode falls back to the broadcast path with same result
    torch.use_deterministic_algorithms(True)
    try:
        assert torch.equal(_bincount(x, minlength=4), torch.tensor([1, 2, 3, 0]))
    finally:
        torch.use_deterministic_algorithms(False)


def test_cumsum_deterministic_path():
    x = torch.randn(100)
    assert torch.allclose(_cumsum(x, dim=0), torch.cumsum(x, dim=0))


def test_interp_matches_numpy():
    x = torch.linspace(0, 1, 11)
    xp = torch.tensor([0.0, 0.5, 1.0])
    fp = torch.tensor([0.0, 2.0, 1.0])
    ref = np.interp(x.numpy(), xp.numpy(), fp.numpy())
    assert torch.allclose(interp(x, xp, fp), torch.from_numpy(ref).float(), atol=1e-6)


def test_safe_helpers():
    assert _safe_divide(torch.tensor(1.0), torch.tensor(0.0)) == 0.0
    assert _safe_divide(torch.tensor(1.0), torch.tensor(0.0), zero_division=5.0) == 5.0
    assert torch.isfinite(_safe_xlogy(torch.tensor(0.0), torch.tensor(0.0)))
    a, b = torch.randn(4, 3, dtype=torch.half), torch.randn(3, 5, dtype=torch.half)
    assert _safe_matmul(a, b).shape == (4, 5)


def test_auc_and_reduce():
    x = torch.tensor([0.0, 0.5, 1.0])
    y = torch.tensor([0.0, 0.5, 1.0])
    assert abs(float(_auc_compute_without_check(x, y, 1.0)) - 0.5) < 1e-6
    t = torch.tensor([[1.0, 2.0], [3.0, 4.0]])
    assert reduce(t, "sum") == 10
    assert reduce(t, "elementwise_mean") == 2.5
    assert torch.equal(reduce(t, "none"), t)
    with pytest.raises(ValueError):
        reduce(t, "bogus")
    num = torch.tensor([1.0, 2.0])
    denom = torch.tensor([2.0, 2.0])
    weights = torch.tensor([1.0, 1.0])
    assert torch.allclose(class_reduce(num, denom, weights, "micro"), torch.tensor(0.75))


def test_check_same_shape_raises():
    with pytest.raises(RuntimeError, match="same shape"):
        _check_same_shape(torch.zeros(2), torch.zeros(3))
