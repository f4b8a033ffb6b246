"""Image metrics: UQI, SAM, ERGAS, total variation, RMSE-SW, SCC, RASE, VIF.

Parity: torchmetrics ``functional/image/{uqi,sam,ergas,tv,rmse_sw,scc,rase,vif}.py``.
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple, Union

import torch
import torch.nn.functional as F
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape
from metrics_amd.utilities.distributed import reduce
from metrics_amd.functional.image.ssim import _gaussian_kernel_2d


def _image_check(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    if preds.dtype != target.dtype:
        target = target.to(preds.dtype)
    _check_same_shape(preds, target)
    if preds.ndim != 4:
        raise ValueError(f"Expected `preds` and `target` to have BxCxHxW shape. Got preds: {preds.shape}.")
    return preds, target


def universal_image_quality_index(
    preds: Tensor,
    target: Tensor,
    kernel_size: Sequence[int] = (11, 11),
    sigma: Sequence[float] = (1.5, 1.5),
    reduction: Optional[str] = "elementwise_mean",
) -> Tensor:
    """Universal image quality index (SSIM with C1 = C2 = 0)."""
    preds, target = _image_check(preds, target)
    channel = preds.size(1)
    kernel = _gaussian_kernel_2d(channel, kernel_size, sigma, preds.dtype, preds.device)
    pad_h = (kernel_size[0] - 1) // 2
    pad_w = (kernel_size[1] - 1) // 2
    preds_p = F.pad(preds, (pad_w, pad_w, pad_h, pad_h), mode="reflect")
    target_p = F.pad(target, (pad_w, pad_w, pad_h, pad_h), mode="reflect")

    input_list = torch.cat((preds_p, target_p, preds_p * preds_p, target_p * target_p, preds_p * target_p))
    outputs = F.conv2d(input_list, kernel, groups=channel)
    output_list = outputs.split(preds.shape[0])

    mu_pred_sq = output_list[0].pow(2)
    mu_target_sq = output_list[1].pow(2)
    mu_pred_target = output_list[0] * output_list[1]

    sigma_pred_sq = output_list[2] - mu_pred_sq
    sigma_target_sq = output_list[3] - mu_target_sq
    sigma_pred_target = output_list[4] - mu_pred_target

    upper = 2 * sigma_pred_target
    lower = sigma_pred_sq + sigma_target_sq
    eps = torch.finfo(preds.dtype).eps
    uqi_idx = ((2 * mu_pred_target) * upper) / ((mu_pred_sq + mu_target_sq) * lower + eps)
    uqi_idx = uqi_idx[..., pad_h:-pad_h, pad_w:-pad_w]
    # the reduction applies to the raw quality MAP (reference uqi.py):
    # 'none' returns the (B,C,H',W') map, 'sum' sums every element
    return reduce(uqi_idx, reduction or "none")


def spectral_angle_mapper(
    preds: Tensor,
    target: Tensor,
    reduction: Optional[str] = "elementwise_mean",
) -> Tensor:
    """Spectral angle (radians) between pred/target spectra at each pixel."""
    preds, target = _image_check(preds, target)
    if preds.shape[1] <= 1:
        raise ValueError(f"Expected channel dimension of `preds` and `target` to be larger than 1. Got preds: {preds.shape[1]}.")
    dot_product = (preds * target).sum(dim=1)
    preds_norm = preds.norm(dim=1)
    target_norm = target.norm(dim=1)
    sam_score = torch.clamp(dot_product / (preds_norm * target_norm), -1, 1).acos()
    return reduce(sam_score, reduction or "none")


def error_relative_global_dimensionless_synthesis(
    preds: Tensor,
    target: Tensor,
    ratio: float = 4,
    reduction: Optional[str] = "elementwise_mean",
) -> Tensor:
    """ERGAS for pan-sharpening quality."""
    preds, target = _image_check(preds, target)
    b, c, h, w = preds.shape
    preds = preds.reshape(b, c, h * w)
    target = target.reshape(b, c, h * w)

    diff = preds - target
    sum_squared_error = torch.sum(diff * diff, dim=2)
    rmse_per_band = torch.sqrt(sum_squared_error / (h * w))
    mean_target = torch.mean(target, dim=2)

    ergas_score = 100 / ratio * torch.sqrt(torch.sum(rmse_per_band**2 / mean_target**2, dim=1) / c)
    return reduce(ergas_score, reduction or "none")


def _pad1(t: Tensor, dim: int, pad: int, outer_pad: int) -> Tensor:
    # scipy-style symmetric padding (edge pixel repeated), size-preserving for a
    # valid w-wide box filter: left pad `pad`, right pad `pad + outer_pad - 1`.
    n = t.shape[dim]
    left = t.index_select(dim, torch.arange(pad - 1, -1, -1, device=t.device))
    right = t.index_select(dim, torch.arange(n - 1, n - pad - outer_pad, -1, device=t.device))
    return torch.cat((left, t, right), dim)


def _scipy_uniform_filter(x: Tensor, window_size: int) -> Tensor:
    """Size-preserving box filter matching scipy.ndimage.uniform_filter(mode='reflect').

    Parity: reference functional/image/utils.py:113 (_uniform_filter).
    """
    pad, outer = window_size // 2, window_size % 2
    x = _pad1(_pad1(x, 2, pad, outer), 3, pad, outer)
    c = x.shape[1]
    kernel = torch.ones(c, 1, window_size, window_size, dtype=x.dtype, device=x.device) / (window_size**2)
    return F.conv2d(x, kernel, groups=c)


def relative_average_spectral_error(preds: Tensor, target: Tensor, window_size: int = 8) -> Tensor:
    """RASE: relative average spectral error using sliding-window RMSE.

    Parity: reference functional/image/rase.py (_rase_update/_rase_compute).
    """
    preds, target = _image_check(preds, target)
    _, rmse_map = _rmse_sw_maps(preds, target, window_size)
    rmse_map = rmse_map.sum(0) / preds.shape[0]  # mean over images -> (C,H,W)
    target_mean = (_scipy_uniform_filter(target, window_size) / (window_size**2)).sum(0) / preds.shape[0]
    target_mean = target_mean.mean(0)  # mean over channels -> (H,W)
    rase_map = 100 / target_mean * torch.sqrt(torch.mean(rmse_map**2, 0))
    crop = round(window_size / 2)
    return torch.mean(rase_map[crop:-crop, crop:-crop])


def _rmse_sw_maps(preds: Tensor, target: Tensor, window_size: int) -> Tuple[Tensor, Tensor]:
    """Per-pixel sliding-window RMSE map (size-preserving) + edge-cropped mean."""
    rmse_map = torch.sqrt(_scipy_uniform_filter((preds - target) ** 2, window_size))
    crop = round(window_size / 2)
    rmse_mean = rmse_map[..., crop:-crop, crop:-crop].mean()
    return rmse_mean, rmse_map


def root_mean_squared_error_using_sliding_window(
    preds: Tensor, target: Tensor, window_size: int = 8, return_rmse_map: bool = False
):
    """RMSE averaged over sliding windows (edges cropped by round(w/2) like the reference)."""
    preds, target = _image_check(preds, target)
    if not isinstance(window_size, int) or window_size < 1:
        raise ValueError(f"Argument `window_size` is expected to be a positive integer, but got {window_size}")
    rmse_mean, rmse_map = _rmse_sw_maps(preds, target, window_size)
    if return_rmse_map:
        return rmse_mean, rmse_map
    return rmse_mean


def total_variation(img: Tensor, reduction: Optional[str] = "sum") -> Tensor:
    """Total variation: sum of absolute spatial gradients."""
    if img.ndim != 4:
        raise RuntimeError(f"Expected input `img` to be an 4D tensor, but got {img.shape}")
    diff1 = img[..., 1:, :] - img[..., :-1, :]
    diff2 = img[..., :, 1:] - img[..., :, :-1]
    res1 = diff1.abs().sum([1, 2, 3])
    res2 = diff2.abs().sum([1, 2, 3])
    score = res1 + res2
    if reduction == "mean":
        return score.mean()
    if reduction == "sum":
        return score.sum()
    if reduction is None or reduction == "none":
        return score
    raise ValueError("Expected argument `reduction` to either be 'sum', 'mean', 'none' or None")


def _symmetric_pad2d(x: Tensor, left: int, right: int, top: int, bottom: int) -> Tensor:
    """Symmetric padding (``d c b a | a b c d | d c b a``) — edge pixel repeated.

    torch's ``reflect`` mode excludes the edge pixel, so it cannot be used here
    (reference functional/image/scc.py ``_symmetric_reflect_pad_2d``).
    """
    parts_w = [x[:, :, :, :left].flip(dims=[3]), x, x[:, :, :, x.shape[3] - right :].flip(dims=[3])]
    x = torch.cat([p for p in parts_w if p.shape[3] > 0], dim=3)
    parts_h = [x[:, :, :top, :].flip(dims=[2]), x, x[:, :, x.shape[2] - bottom :, :].flip(dims=[2])]
    return torch.cat([p for p in parts_h if p.shape[2] > 0], dim=2)


def spatial_correlation_coefficient(
    preds: Tensor,
    target: Tensor,
    hp_filter: Optional[Tensor] = None,
    window_size: int = 8,
    reduction: Optional[str] = "mean",
) -> Tensor:
    """Spatial correlation coefficient after high-pass (Laplacian) filtering.

    Matches the reference/sewar pipeline (reference functional/image/scc.py):
    signal-convolved doubled Laplacian with symmetric padding, then stride-1
    zero-padded window statistics, correlation per pixel, mean per sample.
    """
    if preds.ndim == 3:
        preds = preds.unsqueeze(1)
        target = target.unsqueeze(1)
    if target.dtype != preds.dtype:
        target = target.to(preds.dtype)
    preds, target = _image_check(preds.float(), target.float())
    if not window_size > 0:
        raise ValueError(f"Expected `window_size` to be a positive integer. Got {window_size}.")
    if window_size > preds.size(2) or window_size > preds.size(3):
        raise ValueError(
            f"Expected `window_size` to be less than or equal to the size of the image."
            f" Got window_size: {window_size} and image size: {preds.size(2)}x{preds.size(3)}."
        )
    if reduction is None:
        reduction = "none"
    if reduction not in ("mean", "none"):
        raise ValueError(f"Expected reduction to be 'mean' or 'none', but got {reduction}")
    if hp_filter is None:
        hp_filter = torch.tensor([[-1.0, -1.0, -1.0], [-1.0, 8.0, -1.0], [-1.0, -1.0, -1.0]])
    c = preds.shape[1]
    kern = hp_filter.to(preds).flip([0, 1]).expand(c, 1, *hp_filter.shape)

    # doubled Laplacian via signal convolution (flipped kernel, symmetric pad)
    kh, kw = hp_filter.shape
    lw, rw = (kw - 1) // 2, kw - 1 - (kw - 1) // 2
    th, bh = (kh - 1) // 2, kh - 1 - (kh - 1) // 2
    hp_preds = F.conv2d(_symmetric_pad2d(preds, lw, rw, th, bh), kern, groups=c) * 2.0
    hp_target = F.conv2d(_symmetric_pad2d(target, lw, rw, th, bh), kern, groups=c) * 2.0

    # stride-1 window statistics with zero padding (ceil left/top, floor right/bottom)
    lp = (window_size - 1 + 1) // 2
    rp = (window_size - 1) // 2
    win = torch.ones(c, 1, window_size, window_size, device=preds.device, dtype=preds.dtype) / window_size**2
    xp = F.pad(hp_preds, (lp, rp, lp, rp))
    yp = F.pad(hp_target, (lp, rp, lp, rp))
    mu_x = F.conv2d(xp, win, groups=c)
    mu_y = F.conv2d(yp, win, groups=c)
    var_x = F.conv2d(xp**2, win, groups=c) - mu_x**2
    var_y = F.conv2d(yp**2, win, groups=c) - mu_y**2
    cov = F.conv2d(yp * xp, win, groups=c) - mu_y * mu_x

    denom = torch.sqrt(var_y.clamp(min=0)) * torch.sqrt(var_x.clamp(min=0))
    scc = torch.where(denom > 0, cov / torch.where(denom > 0, denom, torch.ones_like(denom)), torch.zeros_like(cov))
    if reduction == "none":
        return scc.mean(dim=(1, 2, 3))
    return scc.mean()


def visual_information_fidelity(preds: Tensor, target: Tensor, sigma_n_sq: float = 2.0) -> Tensor:
    """VIF-P (pixel-domain visual information fidelity), averaged over batch."""
    preds, target = _image_check(preds.float(), target.float())
    if preds.shape[-1] < 41 or preds.shape[-2] < 41:
        raise ValueError("Invalid size of preds. Expected at least 41x41")
    eps = torch.finfo(preds.dtype).eps
    b, c = preds.shape[:2]
    preds = preds.reshape(b * c, 1, *preds.shape[2:])
    target = target.reshape(b * c, 1, *target.shape[2:])

    num = torch.zeros(b * c, device=preds.device)
    den = torch.zeros(b * c, device=preds.device)
    for scale in range(1, 5):
        n = 2 ** (4 - scale + 1) + 1
        sd = n / 5.0
        kernel = _gaussian_kernel_2d(1, (n, n), (sd, sd), preds.dtype, preds.device)
        if scale > 1:
            target = F.conv2d(target, kernel)[:, :, ::2, ::2]
            preds = F.conv2d(preds, kernel)[:, :, ::2, ::2]

        mu1 = F.conv2d(target, kernel)
        mu2 = F.conv2d(preds, kernel)
        mu1_sq, mu2_sq, mu1_mu2 = mu1 * mu1, mu2 * mu2, mu1 * mu2
        sigma1_sq = F.conv2d(target * target, kernel) - mu1_sq
        sigma2_sq = F.conv2d(preds * preds, kernel) - mu2_sq
        sigma12 = F.conv2d(target * preds, kernel) - mu1_mu2

        sigma1_sq = sigma1_sq.clamp(min=0)
        sigma2_sq = sigma2_sq.clamp(min=0)

        g = sigma12 / (sigma1_sq + eps)
        sv_sq = sigma2_sq - g * sigma12

        g = torch.where(sigma1_sq >= eps, g, torch.zeros_like(g))
        sv_sq = torch.where(sigma1_sq >= eps, sv_sq, sigma2_sq)
        sigma1_sq = torch.where(sigma1_sq >= eps, sigma1_sq, torch.zeros_like(sigma1_sq))

        g = torch.where(sigma2_sq >= eps, g, torch.zeros_like(g))
        sv_sq = torch.where(sigma2_sq >= eps, sv_sq, torch.zeros_like(sv_sq))

        sv_sq = torch.where(g >= 0, sv_sq, sigma2_sq)
        g = g.clamp(min=0)
        sv_sq = sv_sq.clamp(min=eps)

        num += torch.sum(torch.log10(1.0 + (g**2.0) * sigma1_sq / (sv_sq + sigma_n_sq)), dim=(1, 2, 3))
        den += torch.sum(torch.log10(1.0 + sigma1_sq / sigma_n_sq), dim=(1, 2, 3))

    return (num / den).reshape(b, c).mean()
