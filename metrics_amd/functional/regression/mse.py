"""MSE. Parity: torchmetrics ``functional/regression/mse.py``.

GPU single-output path: one fused deterministic HIP reduction
(ops.err_reduce_sum) instead of sub->pow->sum kernel chains.
"""
from __future__ import annotations

from typing import Tuple, Union

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.checks import _check_same_shape


def _mean_squared_error_update(preds: Tensor, target: Tensor, num_outputs: int) -> Tuple[Tensor, int]:
    """Return (sum of squared errors, n_obs)."""
    _check_same_shape(preds, target)
    if num_outputs == 1:
        preds = preds.view(-1)
        target = target.view(-1)
        if preds.is_cuda and preds.dtype in (torch.float32, torch.bfloat16) and target.dtype == preds.dtype:
            sse = ops.err_reduce_sum(preds, target, "sq_err")[0].to(torch.float32 if preds.dtype != torch.float64 else preds.dtype)
            return sse, target.shape[0]
    diff = preds - target
    sum_squared_error = torch.sum(diff * diff, dim=0)
    return sum_squared_error, target.shape[0]


def _mean_squared_error_compute(sum_squared_error: Tensor, num_obs: Union[int, Tensor], squared: bool = True) -> Tensor:
    return sum_squared_error / num_obs if squared else torch.sqrt(sum_squared_error / num_obs)


def mean_squared_error(preds: Tensor, target: Tensor, squared: bool = True, num_outputs: int = 1) -> Tensor:
    """Mean squared error (or RMSE if ``squared=False``)."""
    sum_squared_error, num_obs = _mean_squared_error_update(preds, target, num_outputs)
    return _mean_squared_error_compute(sum_squared_error, num_obs, squared)
