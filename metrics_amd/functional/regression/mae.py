"""MAE. Parity: torchmetrics ``functional/regression/mae.py``."""
from __future__ import annotations

from typing import Tuple, Union

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.checks import _check_same_shape


def _mean_absolute_error_update(preds: Tensor, target: Tensor, num_outputs: int = 1) -> Tuple[Tensor, int]:
    _check_same_shape(preds, target)
    if num_outputs == 1:
        p = preds.reshape(-1)
        t = target.reshape(-1)
        if p.is_cuda and p.dtype in (torch.float32, torch.bfloat16) and t.dtype == p.dtype:
            return ops.err_reduce_sum(p, t, "abs_err")[0].float(), t.numel()
        preds, target = p, t
    else:
        preds = preds.view(-1, num_outputs) if preds.ndim > 1 else preds
        target = target.view(-1, num_outputs) if target.ndim > 1 else target
    preds = preds if preds.is_floating_point() else preds.float()
    target = target if target.is_floating_point() else target.float()
    sum_abs_error = torch.sum(torch.abs(preds - target), dim=0)
    return sum_abs_error, target.shape[0]


def _mean_absolute_error_compute(sum_abs_error: Tensor, num_obs: Union[int, Tensor]) -> Tensor:
    return sum_abs_error / num_obs


def mean_absolute_error(preds: Tensor, target: Tensor, num_outputs: int = 1) -> Tensor:
    """Mean absolute error."""
    sum_abs_error, num_obs = _mean_absolute_error_update(preds, target, num_outputs)
    return _mean_absolute_error_compute(sum_abs_error, num_obs)
