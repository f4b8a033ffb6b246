"""MSLE. Parity: torchmetrics ``functional/regression/log_mse.py``."""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.checks import _check_same_shape


def _mean_squared_log_error_update(preds: Tensor, target: Tensor) -> Tuple[Tensor, int]:
    _check_same_shape(preds, target)
    p = preds.reshape(-1)
    t = target.reshape(-1)
    if p.is_cuda and p.dtype in (torch.float32, torch.bfloat16) and t.dtype == p.dtype:
        return ops.err_reduce_sum(p, t, "sq_log_err")[0].float(), t.numel()
    sum_squared_log_error = torch.sum(torch.pow(torch.log1p(p) - torch.log1p(t), 2))
    return sum_squared_log_error, t.numel()


def _mean_squared_log_error_compute(sum_squared_log_error: Tensor, num_obs: int) -> Tensor:
    return sum_squared_log_error / num_obs


def mean_squared_log_error(preds: Tensor, target: Tensor) -> Tensor:
    """Mean squared logarithmic error."""
    s, n = _mean_squared_log_error_update(preds, target)
    return _mean_squared_log_error_compute(s, n)
