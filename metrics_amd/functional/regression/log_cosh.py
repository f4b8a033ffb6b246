"""LogCosh error. Parity: torchmetrics ``functional/regression/log_cosh.py``."""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.checks import _check_same_shape


def _unsqueeze_tensors(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    if preds.ndim == 2:
        return preds, target
    return preds.unsqueeze(1), target.unsqueeze(1)


def _log_cosh_error_update(preds: Tensor, target: Tensor, num_outputs: int) -> Tuple[Tensor, int]:
    _check_same_shape(preds, target)
    if num_outputs == 1 and preds.is_cuda and preds.dtype in (torch.float32, torch.bfloat16):
        p = preds.reshape(-1)
        t = target.reshape(-1)
        return ops.err_reduce_sum(p, t, "logcosh")[0].float(), t.numel()
    preds, target = _unsqueeze_tensors(preds.reshape(-1, num_outputs), target.reshape(-1, num_outputs))
    diff = preds - target
    # log(cosh(x)) = x + softplus(-2x) - log(2), numerically stable
    sum_log_cosh_error = (diff + torch.nn.functional.softplus(-2 * diff) - torch.log(torch.tensor(2.0, device=diff.device))).sum(0).squeeze(-1)
    return sum_log_cosh_error, preds.shape[0]


def _log_cosh_error_compute(sum_log_cosh_error: Tensor, num_obs: int) -> Tensor:
    return (sum_log_cosh_error / num_obs).squeeze()


def log_cosh_error(preds: Tensor, target: Tensor) -> Tensor:
    """LogCosh error."""
    num_outputs = 1 if preds.ndim == 1 else preds.shape[-1]
    s, n = _log_cosh_error_update(preds, target, num_outputs)
    return _log_cosh_error_compute(s, n)
