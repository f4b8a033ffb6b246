"""MAPE / weighted MAPE / symmetric MAPE.

Parity: torchmetrics ``functional/regression/{mape,weighted_mape,symmetric_mape}.py``.
"""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.checks import _check_same_shape

_EPS = 1.17e-06


def _mean_absolute_percentage_error_update(preds: Tensor, target: Tensor, epsilon: float = _EPS) -> Tuple[Tensor, int]:
    _check_same_shape(preds, target)
    p = preds.reshape(-1)
    t = target.reshape(-1)
    if p.is_cuda and p.dtype in (torch.float32, torch.bfloat16) and t.dtype == p.dtype:
        return ops.err_reduce_sum(p, t, "ape", eps=epsilon)[0].float(), t.numel()
    abs_per_error = torch.abs(p - t) / torch.clamp(torch.abs(t), min=epsilon)
    return torch.sum(abs_per_error), t.numel()


def _mean_absolute_percentage_error_compute(sum_abs_per_error: Tensor, num_obs: int) -> Tensor:
    return sum_abs_per_error / num_obs


def mean_absolute_percentage_error(preds: Tensor, target: Tensor) -> Tensor:
    """Mean absolute percentage error."""
    s, n = _mean_absolute_percentage_error_update(preds, target)
    return _mean_absolute_percentage_error_compute(s, n)


def _weighted_mean_absolute_percentage_error_update(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    _check_same_shape(preds, target)
    preds = preds.reshape(-1)
    target = target.reshape(-1)
    sum_abs_error = (preds - target).abs().sum()
    sum_scale = target.abs().sum()
    return sum_abs_error, sum_scale


def _weighted_mean_absolute_percentage_error_compute(
    sum_abs_error: Tensor, sum_scale: Tensor, epsilon: float = _EPS
) -> Tensor:
    return sum_abs_error / torch.clamp(sum_scale, min=epsilon)


def weighted_mean_absolute_percentage_error(preds: Tensor, target: Tensor) -> Tensor:
    """Weighted MAPE."""
    sum_abs_error, sum_scale = _weighted_mean_absolute_percentage_error_update(preds, target)
    return _weighted_mean_absolute_percentage_error_compute(sum_abs_error, sum_scale)


def _symmetric_mean_absolute_percentage_error_update(
    preds: Tensor, target: Tensor, epsilon: float = _EPS
) -> Tuple[Tensor, int]:
    _check_same_shape(preds, target)
    preds = preds.reshape(-1)
    target = target.reshape(-1)
    abs_per_error = 2 * torch.abs(preds - target) / torch.clamp(torch.abs(target) + torch.abs(preds), min=epsilon)
    return torch.sum(abs_per_error), target.numel()


def _symmetric_mean_absolute_percentage_error_compute(sum_abs_per_error: Tensor, num_obs: int) -> Tensor:
    return sum_abs_per_error / num_obs


def symmetric_mean_absolute_percentage_error(preds: Tensor, target: Tensor) -> Tensor:
    """Symmetric MAPE."""
    s, n = _symmetric_mean_absolute_percentage_error_update(preds, target)
    return _symmetric_mean_absolute_percentage_error_compute(s, n)
