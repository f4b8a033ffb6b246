"""PSNR (+ PSNRB). Parity: torchmetrics ``functional/image/{psnr,psnrb}.py``."""
from __future__ import annotations

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
    """Mean boundary-vs-interior squared difference factor used by PSNRB."""
    _, _, h, w = x.shape
    h_blocks, w_blocks = h // block_size, w // block_size

    h_bound = torch.arange(block_size - 1, block_size * h_blocks - 1, block_size, device=x.device)
    w_bound = torch.arange(block_size - 1, block_size * w_blocks - 1, block_size, device=x.device)

    d_b_h = (x[:, :, h_bound, :] - x[:, :, h_bound + 1, :]).pow(2).sum(dim=(1, 2, 3))
    d_b_w = (x[:, :, :, w_bound] - x[:, :, :, w_bound + 1]).pow(2).sum(dim=(1, 2, 3))

    all_h = torch.arange(0, h - 1, device=x.device)
    all_w = torch.arange(0, w - 1, device=x.device)
    nonb_h = all_h[~torch.isin(all_h, h_bound)]
    nonb_w = all_w[~torch.isin(all_w, w_bound)]

    d_bc_h = (x[:, :, nonb_h, :] - x[:, :, nonb_h + 1, :]).pow(2).sum(dim=(1, 2, 3))
    d_bc_w = (x[:, :, :, nonb_w] - x[:, :, :, nonb_w + 1]).pow(2).sum(dim=(1, 2, 3))

    n_b = x.shape[1] * (w * len(h_bound) + h * len(w_bound))
    n_bc = x.shape[1] * (w * len(nonb_h) + h * len(nonb_w))

    d_b = (d_b_h + d_b_w) / n_b
    d_bc = (d_bc_h + d_bc_w) / n_bc
    t = torch.log2(torch.tensor(block_size, device=x.device).float()) / torch.log2(
        torch.tensor(min(h, w), device=x.device).float()
    )
    return torch.where(d_b > d_bc, t * (d_b - d_bc), torch.zeros_like(d_b))


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

    mse = ((preds - target) ** 2).mean(dim=(1, 2, 3))
    mse_b = mse + bef
    return (10 * torch.log10(data_range**2 / mse_b)).mean()
