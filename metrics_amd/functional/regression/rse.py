"""Relative squared error. Parity: torchmetrics ``functional/regression/rse.py``."""
from __future__ import annotations

from typing import Union

import torch
from torch import Tensor

from metrics_amd.functional.regression.r2 import _r2_score_update


def _relative_squared_error_compute(
    sum_squared_obs: Tensor,
    sum_obs: Tensor,
    sum_squared_error: Tensor,
    num_obs: Union[int, Tensor],
    squared: bool = True,
) -> Tensor:
    epsilon = torch.finfo(sum_squared_error.dtype).eps
    rse = sum_squared_error / torch.clamp(sum_squared_obs - sum_obs * sum_obs / num_obs, min=epsilon)
    if not squared:
        rse = torch.sqrt(rse)
    return torch.mean(rse)


def relative_squared_error(preds: Tensor, target: Tensor, squared: bool = True) -> Tensor:
    """Relative squared error Σ(y-ŷ)² / Σ(y-ȳ)² (RRSE if squared=False)."""
    sum_squared_obs, sum_obs, rss, num_obs = _r2_score_update(preds, target)
    return _relative_squared_error_compute(sum_squared_obs, sum_obs, rss, num_obs, squared=squared)
