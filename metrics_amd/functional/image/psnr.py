"""PSNR (+ PSNRB). Parity: torchmetrics ``functional/image/{psnr,psnrb}.py``."""
from __future__ import annotations

import math
from typing import Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape


def _psnr_update(
    preds: Tensor,
    target: Tensor,
    dim: Optional[Union[int, Tuple[int, ...]]] = None,
) -> Tuple[Tensor, Tensor]:
    if dim is None:
        sum_squared_error = torch.sum(torch.pow(preds - target, 2))
        num_obs = torch.tensor(target.numel(), device=target.device)
        return sum_squared_error, num_obs

    diff = preds - target
    sum_squared_error = torch.sum(diff * diff, dim=dim)
    dim_list = [dim] if isinstance(dim, int) else list(dim)
    if not dim_list:
        num_obs = torch.tensor(target.numel(), device=target.device)
    else:
        num_obs = torch.tensor(target.size(), device=target.device)[dim_list].prod()
        num_obs = num_obs.expand_as(sum_squared_error)
    return sum_squared_error, num_obs


def _psnr_compute(
    sum_squared_error: Tensor,
    num_obs: Tensor,
    data_range: Tensor,
    base: float = 10.0,
    reduction: str = "elementwise_mean",
) -> Tensor:
    psnr_base_e = 2 * torch.log(data_range) - torch.log(sum_squared_error / num_obs)
    psnr_vals = psnr_base_e * (10 / torch.log(torch.tensor(base)))
    from metrics_amd.utilities.distributed import reduce

    return reduce(psnr_vals, reduction=reduction)


def peak_signal_noise_ratio(
    preds: Tensor,
    target: Tensor,
    data_range: Optional[Union[float, Tuple[float, float]]] = None,
    base: float = 10.0,
    reduction: str = "elementwise_mean",
    dim: Optional[Union[int, Tuple[int, ...]]] = None,
) -> Tensor:
    """Peak signal-to-noise ratio."""
    _check_same_shape(preds, target)
    if dim is None and reduction != "elementwise_mean":
        import warnings

        warnings.warn(f"The `reduction={reduction}` will not have any effect when `dim` is None.", stacklevel=2)

    if data_range is None:
        if dim is not None:
            raise ValueError("The `data_range` must be given when `dim` is not None.")
        data_range_t = target.max() - target.min()
    elif isinstance(data_range, tuple):
        preds = torch.clamp(preds, min=data_range[0], max=data_range[1])
        target = torch.clamp(target, min=data_range[0], max=data_range[1])
        data_range_t = torch.tensor(data_range[1] - data_range[0], device=target.device)
    else:
        data_range_t = torch.tensor(float(data_range), device=target.device)
    sum_squared_error, num_obs = _psnr_update(preds, target, dim=dim)
    return _psnr_compute(sum_squared_error, num_obs, data_range_t, base=base, reduction=reduction)


def _blocking_effect_factor(x: Tensor, block_size: int = 8) -> Tensor:
    """Boundary-vs-interior squared-difference factor used by PSNRB.

    Matches the reference's counting exactly (reference functional/image/psnrb.py
    ``_compute_bef``): the "boundary" columns/rows are every ``block_size``-th
    difference, normalizers use the reference's ``H * (W / block) - 1`` form, and
    one scalar is produced for the whole (grayscale) batch.
    """
    _, channels, height, width = x.shape
    if channels > 1:
        raise ValueError(f"`psnrb` metric expects grayscale images, but got images with {channels} channels.")

    h_bound = torch.arange(block_size - 1, width - 1, block_size, device=x.device)
    all_w = torch.arange(width - 1, device=x.device)
    h_nonb = all_w[~torch.isin(all_w, h_bound)]
    v_bound = torch.arange(block_size - 1, height - 1, block_size, device=x.device)
    all_h = torch.arange(height - 1, device=x.device)
    v_nonb = all_h[~torch.isin(all_h, v_bound)]

    d_b = (x[:, :, :, h_bound] - x[:, :, :, h_bound + 1]).pow(2).sum()
    d_bc = (x[:, :, :, h_nonb] - x[:, :, :, h_nonb + 1]).pow(2).sum()
    d_b = d_b + (x[:, :, v_bound, :] - x[:, :, v_bound + 1, :]).pow(2).sum()
    d_bc = d_bc + (x[:, :, v_nonb, :] - x[:, :, v_nonb + 1, :]).pow(2).sum()

    n_hb = height * (width / block_size) - 1
    n_hbc = height * (width - 1) - n_hb
    n_vb = width * (height / block_size) - 1
    n_vbc = width * (height - 1) - n_vb
    d_b = d_b / (n_hb + n_vb)
    d_bc = d_bc / (n_hbc + n_vbc)
    t = math.log2(block_size) / math.log2(min(height, width)) if d_b > d_bc else 0
    return t * (d_b - d_bc)


def peak_signal_noise_ratio_with_blocked_effect(
    preds: Tensor,
    target: Tensor,
    block_size: int = 8,
) -> Tensor:
    """PSNR-B: PSNR penalized by the blocking effect factor."""
    _check_same_shape(preds, target)
    if preds.ndim != 4:
        raise ValueError(f"Expected 4D (N,C,H,W) input, got {preds.ndim}D")
    data_range = target.max() - target.min()
    bef = _blocking_effect_factor(preds, block_size=block_size)

    mse_b = ((preds - target) ** 2).sum() / target.numel() + bef
    if data_range > 2:
        return 10 * torch.log10(data_range**2 / mse_b)
    return 10 * torch.log10(1.0 / mse_b)
