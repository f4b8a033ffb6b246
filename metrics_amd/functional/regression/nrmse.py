"""Normalized RMSE. Parity: torchmetrics ``functional/regression/nrmse.py``."""
from __future__ import annotations

from typing import Tuple, Union

import torch
from torch import Tensor

from metrics_amd.functional.regression.mse import _mean_squared_error_update


def _normalized_root_mean_squared_error_update(
    preds: Tensor,
    target: Tensor,
    num_outputs: int,
    normalization: str = "mean",
) -> Tuple[Tensor, int, Tensor]:
    """Return (sum sq error, n, the denominator statistic for this batch)."""
    sum_squared_error, num_obs = _mean_squared_error_update(preds, target, num_outputs)

    target = target.view(-1) if num_outputs == 1 else target
    if normalization == "mean":
        denom = target.mean(0)
    elif normalization == "range":
        denom = target.max(0).values - target.min(0).values
    elif normalization == "std":
        denom = target.std(0, correction=0)
    elif normalization == "l2":
        denom = target.norm(2, dim=0)
    else:
        raise ValueError(
            f"Argument `normalization` should be either 'mean', 'range', 'std' or 'l2', but got {normalization}"
        )
    return sum_squared_error, num_obs, denom


def _normalized_root_mean_squared_error_compute(
    sum_squared_error: Tensor, num_obs: Union[int, Tensor], denom: Tensor
) -> Tensor:
    rmse = torch.sqrt(sum_squared_error / num_obs)
    # the reference divides by the RAW statistic: a negative target mean
    # legitimately yields a negative score
    return rmse / denom


def normalized_root_mean_squared_error(
    preds: Tensor, target: Tensor, normalization: str = "mean", num_outputs: int = 1
) -> Tensor:
    """RMSE normalized by mean / range / std / L2-norm of the target."""
    sum_squared_error, num_obs, denom = _normalized_root_mean_squared_error_update(
        preds, target, num_outputs, normalization
    )
    return _normalized_root_mean_squared_error_compute(sum_squared_error, num_obs, denom)
