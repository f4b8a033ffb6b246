"""SSIM / MS-SSIM. Parity: torchmetrics ``functional/image/ssim.py``.

Gaussian/uniform windows as separable grouped convs (MIOpen on ROCm).
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple, Union

import torch
import torch.nn.functional as F
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape


def _gaussian(kernel_size: int, sigma: float, dtype: torch.dtype, device: torch.device) -> Tensor:
    dist = torch.arange((1 - kernel_size) / 2, (1 + kernel_size) / 2, step=1, dtype=dtype, device=device)
    gauss = torch.exp(-torch.pow(dist / sigma, 2) / 2)
    return (gauss / gauss.sum()).unsqueeze(dim=0)


def _gaussian_kernel_2d(
    channel: int, kernel_size: Sequence[int], sigma: Sequence[float], dtype: torch.dtype, device: torch.device
) -> Tensor:
    gaussian_kernel_x = _gaussian(kernel_size[0], sigma[0], dtype, device)
    gaussian_kernel_y = _gaussian(kernel_size[1], sigma[1], dtype, device)
    kernel = torch.matmul(gaussian_kernel_x.t(), gaussian_kernel_y)
    return kernel.expand(channel, 1, kernel_size[0], kernel_size[1])


def _uniform_kernel_2d(
    channel: int, kernel_size: Sequence[int], dtype: torch.dtype, device: torch.device
) -> Tensor:
    kernel = torch.ones(kernel_size[0], kernel_size[1], dtype=dtype, device=device) / (
        kernel_size[0] * kernel_size[1]
    )
    return kernel.expand(channel, 1, kernel_size[0], kernel_size[1])


def _gaussian_kernel_3d(
    channel: int, kernel_size: Sequence[int], sigma: Sequence[float], dtype: torch.dtype, device: torch.device
) -> Tensor:
    gx = _gaussian(kernel_size[0], sigma[0], dtype, device)
    gy = _gaussian(kernel_size[1], sigma[1], dtype, device)
    gz = _gaussian(kernel_size[2], sigma[2], dtype, device)
    kernel_xy = torch.matmul(gx.t(), gy)
    kernel = kernel_xy.unsqueeze(-1) * gz.expand(kernel_size[0], kernel_size[1], kernel_size[2])
    return kernel.expand(channel, 1, kernel_size[0], kernel_size[1], kernel_size[2])


def _try_ssim_hip(
    preds: Tensor,
    target: Tensor,
    gaussian_kernel: bool,
    sigma: Sequence[float],
    kernel_size: Sequence[int],
    data_range_t,
    k1: float,
    k2: float,
    is_3d: bool,
    return_full_image: bool,
    return_contrast_sensitivity: bool,
):
    """K8 fused one-kernel SSIM (csrc/ssim.hip) for the GPU 2D path.

    Returns None when this shape/mode needs the torch formulation (3D,
    full-image output, oversized windows, reflect-pad-invalid sizes).
    """
    if not preds.is_cuda or is_3d or return_full_image:
        return None
    gauss_kernel_size = [int(3.5 * s + 0.5) * 2 + 1 for s in sigma]
    if gaussian_kernel:
        pad_h = (gauss_kernel_size[0] - 1) // 2
        pad_w = (gauss_kernel_size[1] - 1) // 2
    else:
        pad_h = (kernel_size[0] - 1) // 2
        pad_w = (kernel_size[1] - 1) // 2
    B, C, H, W = preds.shape
    if pad_h >= H or pad_w >= W:  # reflect pad invalid: torch path raises
        return None
    if return_contrast_sensitivity and (H <= 2 * pad_h or W <= 2 * pad_w):
        return None
    dtype = preds.dtype
    device = preds.device
    if gaussian_kernel:
        wh = _gaussian(gauss_kernel_size[0], sigma[0], torch.float32, device)[0]
        ww = _gaussian(gauss_kernel_size[1], sigma[1], torch.float32, device)[0]
    else:
        wh = torch.full((kernel_size[0],), 1.0 / kernel_size[0], device=device)
        ww = torch.full((kernel_size[1],), 1.0 / kernel_size[1], device=device)

    if isinstance(data_range_t, Tensor):
        dr = data_range_t.float()
        c1 = c2 = 0.0
    else:
        dr = None
        c1 = float(pow(k1 * data_range_t, 2))
        c2 = float(pow(k2 * data_range_t, 2))
    from metrics_amd.ops import _hip

    try:
        sum_sim, sum_cs = _hip.ssim2d_fused(
            preds, target, wh, ww, c1, c2, dr, k1, k2,
            return_contrast_sensitivity, pad_h, pad_w,
        )
    except _hip.SsimLdsOverflow:
        return None
    sim = (sum_sim / (C * H * W)).to(dtype)
    if return_contrast_sensitivity:
        cs = (sum_cs / (C * (H - 2 * pad_h) * (W - 2 * pad_w))).to(dtype)
        return sim, cs
    return sim


def _ssim_check_inputs(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    if preds.dtype != target.dtype:
        target = target.to(preds.dtype)
    _check_same_shape(preds, target)
    if len(preds.shape) not in (4, 5):
        raise ValueError(f"Expected `preds` and `target` to have BxCxHxW or BxCxDxHxW shape. Got preds: {preds.shape}.")
    return preds, target


def _ssim_compute(
    preds: Tensor,
    target: Tensor,
    gaussian_kernel: bool = True,
    sigma: Union[float, Sequence[float]] = 1.5,
    kernel_size: Union[int, Sequence[int]] = 11,
    data_range: Optional[Union[float, Tuple[float, float]]] = None,
    k1: float = 0.01,
    k2: float = 0.03,
    return_full_image: bool = False,
    return_contrast_sensitivity: bool = False,
):
    is_3d = preds.ndim == 5
    dims = 3 if is_3d else 2

    if not isinstance(kernel_size, Sequence):
        kernel_size = dims * [kernel_size]
    if not isinstance(sigma, Sequence):
        sigma = dims * [sigma]
    if len(kernel_size) != dims or len(sigma) != dims:
        raise ValueError(
            f"`kernel_size` and `sigma` must have {dims} elements for {dims}d input,"
            f" got {kernel_size} and {sigma}"
        )

    if any(x % 2 == 0 or x <= 0 for x in kernel_size):
        raise ValueError(f"Expected `kernel_size` to have odd positive number. Got {kernel_size}.")
    if any(y <= 0 for y in sigma):
        raise ValueError(f"Expected `sigma` to have positive number. Got {sigma}.")

    if data_range is None:
        data_range_t = max(preds.max() - preds.min(), target.max() - target.min())
    elif isinstance(data_range, tuple):
        preds = torch.clamp(preds, min=data_range[0], max=data_range[1])
        target = torch.clamp(target, min=data_range[0], max=data_range[1])
        data_range_t = data_range[1] - data_range[0]
    else:
        data_range_t = data_range

    c1 = pow(k1 * data_range_t, 2)
    c2 = pow(k2 * data_range_t, 2)

    hip_out = _try_ssim_hip(
        preds, target, gaussian_kernel, sigma, kernel_size, data_range_t, k1, k2,
        is_3d, return_full_image, return_contrast_sensitivity,
    )
    if hip_out is not None:
        return hip_out

    channel = preds.size(1)
    dtype = preds.dtype
    device = preds.device
    # with a gaussian window the effective kernel (and padding) size follows
    # sigma, not the kernel_size argument (reference functional/image/ssim.py:126)
    gauss_kernel_size = [int(3.5 * s + 0.5) * 2 + 1 for s in sigma]
    if gaussian_kernel:
        pad_h = (gauss_kernel_size[0] - 1) // 2
        pad_w = (gauss_kernel_size[1] - 1) // 2
    else:
        pad_h = (kernel_size[0] - 1) // 2
        pad_w = (kernel_size[1] - 1) // 2

    if is_3d:
        # reference quirk kept: depth padding always follows kernel_size[2],
        # even for gaussian windows (ref functional/image/ssim.py:136-141)
        pad_d = (kernel_size[2] - 1) // 2
        preds_p = F.pad(preds, (pad_d, pad_d, pad_w, pad_w, pad_h, pad_h), mode="reflect")
        target_p = F.pad(target, (pad_d, pad_d, pad_w, pad_w, pad_h, pad_h), mode="reflect")
        if gaussian_kernel:
            kernel = _gaussian_kernel_3d(channel, gauss_kernel_size, sigma, dtype, device)
        else:
            kernel = torch.ones((channel, 1, *kernel_size), dtype=dtype, device=device) / torch.prod(
                torch.tensor(kernel_size, dtype=dtype, device=device)
            )
    else:
        preds_p = F.pad(preds, (pad_w, pad_w, pad_h, pad_h), mode="reflect")
        target_p = F.pad(target, (pad_w, pad_w, pad_h, pad_h), mode="reflect")
        if gaussian_kernel:
            kernel = _gaussian_kernel_2d(channel, gauss_kernel_size, sigma, dtype, device)
        else:
            kernel = _uniform_kernel_2d(channel, kernel_size, dtype, device)

    input_list = torch.cat((preds_p, target_p, preds_p * preds_p, target_p * target_p, preds_p * target_p))
    outputs = (
        F.conv3d(input_list, kernel, groups=channel) if is_3d else F.conv2d(input_list, kernel, groups=channel)
    )
    output_list = outputs.split(preds.shape[0])

    mu_pred_sq = output_list[0].pow(2)
    mu_target_sq = output_list[1].pow(2)
    mu_pred_target = output_list[0] * output_list[1]

    sigma_pred_sq = torch.clamp(output_list[2] - mu_pred_sq, min=0.0)
    sigma_target_sq = torch.clamp(output_list[3] - mu_target_sq, min=0.0)
    sigma_pred_target = output_list[4] - mu_pred_target

    upper = 2 * sigma_pred_target.to(dtype) + c2
    lower = (sigma_pred_sq + sigma_target_sq).to(dtype) + c2

    # the similarity mean is taken over the FULL map (border ring included);
    # only contrast sensitivity is cropped (reference functional/image/ssim.py:173-186)
    ssim_idx_full_image = ((2 * mu_pred_target + c1) * upper) / ((mu_pred_sq + mu_target_sq + c1) * lower)
    sim = ssim_idx_full_image.reshape(ssim_idx_full_image.shape[0], -1).mean(-1)

    if return_contrast_sensitivity:
        contrast_sensitivity = upper / lower
        if is_3d:
            contrast_sensitivity = contrast_sensitivity[..., pad_h:-pad_h, pad_w:-pad_w, pad_d:-pad_d]
        else:
            contrast_sensitivity = contrast_sensitivity[..., pad_h:-pad_h, pad_w:-pad_w]
        return sim, contrast_sensitivity.reshape(contrast_sensitivity.shape[0], -1).mean(-1)
    if return_full_image:
        return sim, ssim_idx_full_image
    return sim


def structural_similarity_index_measure(
    preds: Tensor,
    target: Tensor,
    gaussian_kernel: bool = True,
    sigma: Union[float, Sequence[float]] = 1.5,
    kernel_size: Union[int, Sequence[int]] = 11,
    reduction: Optional[str] = "elementwise_mean",
    data_range: Optional[Union[float, Tuple[float, float]]] = None,
    k1: float = 0.01,
    k2: float = 0.03,
    return_full_image: bool = False,
    return_contrast_sensitivity: bool = False,
):
    """Structural similarity index measure."""
    preds, target = _ssim_check_inputs(preds, target)
    out = _ssim_compute(
        preds, target, gaussian_kernel, sigma, kernel_size, data_range, k1, k2,
        return_full_image, return_contrast_sensitivity,
    )
    if isinstance(out, tuple):
        similarity, extra = out
    else:
        similarity, extra = out, None

    from metrics_amd.utilities.distributed import reduce

    similarity = reduce(similarity, reduction or "none")
    if extra is not None:
        return similarity, extra
    return similarity


_MS_SSIM_BETAS = (0.0448, 0.2856, 0.3001, 0.2363, 0.1333)


def multiscale_structural_similarity_index_measure(
    preds: Tensor,
    target: Tensor,
    gaussian_kernel: bool = True,
    sigma: Union[float, Sequence[float]] = 1.5,
    kernel_size: Union[int, Sequence[int]] = 11,
    reduction: Optional[str] = "elementwise_mean",
    data_range: Optional[Union[float, Tuple[float, float]]] = None,
    k1: float = 0.01,
    k2: float = 0.03,
    betas: Tuple[float, ...] = _MS_SSIM_BETAS,
    normalize: Optional[str] = "relu",
) -> Tensor:
    """Multi-scale SSIM."""
    preds, target = _ssim_check_inputs(preds, target)
    if not isinstance(betas, tuple) or not all(isinstance(beta, float) for beta in betas):
        raise ValueError("Argument `betas` is expected to be of a tuple of floats")
    if normalize not in ("relu", "simple", None):
        raise ValueError("Argument `normalize` to be expected either `None` or one of 'relu' or 'simple'")

    k0 = kernel_size if isinstance(kernel_size, int) else kernel_size[0]
    k1_ = kernel_size if isinstance(kernel_size, int) else kernel_size[1]
    if preds.size()[-1] < 2 ** len(betas) or preds.size()[-2] < 2 ** len(betas):
        raise ValueError(
            f"For a given number of `betas` parameters {len(betas)}, the image height and width dimensions must be"
            f" larger than or equal to {2 ** len(betas)}."
        )
    _betas_div = max(1, (len(betas) - 1)) ** 2
    if preds.size()[-2] // _betas_div <= k0 - 1:
        raise ValueError(
            f"For a given number of `betas` parameters {len(betas)} and kernel size {k0},"
            f" the image height must be larger than {(k0 - 1) * _betas_div}."
        )
    if preds.size()[-1] // _betas_div <= k1_ - 1:
        raise ValueError(
            f"For a given number of `betas` parameters {len(betas)} and kernel size {k1_},"
            f" the image width must be larger than {(k1_ - 1) * _betas_div}."
        )

    mcs_list: List[Tensor] = []
    sim = None
    is_3d = preds.ndim == 5
    for i in range(len(betas)):
        sim, contrast_sensitivity = _ssim_compute(
            preds, target, gaussian_kernel, sigma, kernel_size, data_range, k1, k2,
            return_contrast_sensitivity=True,
        )
        mcs_list.append(contrast_sensitivity)
        if i < len(betas) - 1:
            if is_3d:
                preds = F.avg_pool3d(preds, (2, 2, 2))
                target = F.avg_pool3d(target, (2, 2, 2))
            else:
                preds = F.avg_pool2d(preds, (2, 2))
                target = F.avg_pool2d(target, (2, 2))

    mcs_list[-1] = sim
    mcs_stack = torch.stack(mcs_list)

    if normalize == "relu":
        mcs_stack = torch.relu(mcs_stack)

    betas_t = torch.tensor(betas, device=mcs_stack.device).view(-1, 1)
    if normalize == "simple":
        mcs_stack = (mcs_stack + 1) / 2
    mcs_weighted = mcs_stack**betas_t
    out = torch.prod(mcs_weighted, axis=0)

    from metrics_amd.utilities.distributed import reduce

    return reduce(out, reduction or "none")
