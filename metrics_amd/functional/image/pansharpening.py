"""Pan-sharpening quality: D_lambda, D_s, QNR, spatial distortion index.

Parity: torchmetrics ``functional/image/{d_lambda,d_s,qnr}.py`` and
``spatial_distortion_index``.
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch
import torch.nn.functional as F
from torch import Tensor

from metrics_amd.functional.image.misc import universal_image_quality_index


def _band_uqi(x: Tensor, y: Tensor, window_size: int = 8) -> Tensor:
    """UQI between two single-band images (N,H,W)."""
    return universal_image_quality_index(
        x.unsqueeze(1), y.unsqueeze(1), kernel_size=(window_size, window_size), sigma=(1.5, 1.5)
    )


def spectral_distortion_index(
    preds: Tensor, target: Tensor, p: int = 1, reduction: str = "elementwise_mean"
) -> Tensor:
    """D_lambda: spectral distortion between fused (preds) and MS (target) images."""
    if preds.ndim != 4 or target.ndim != 4:
        raise ValueError(f"Expected `preds` and `target` to have BxCxHxW shape. Got preds: {preds.shape} and target: {target.shape}.")
    if p <= 0:
        raise ValueError(f"Expected `p` to be a positive integer. Got p: {p}.")
    length = preds.shape[1]
    if length <= 1:
        raise ValueError(f"Expected channel dimension to be larger than 1. Got {length}.")

    m1 = torch.zeros((length, length), device=preds.device)
    m2 = torch.zeros((length, length), device=preds.device)
    for k in range(length):
        for r in range(k, length):
            m1[k, r] = m1[r, k] = _band_uqi(target[:, k], target[:, r]).mean()
            m2[k, r] = m2[r, k] = _band_uqi(preds[:, k], preds[:, r]).mean()
    diff = (m1 - m2).abs() ** p
    # off-diagonal mean
    total = diff.sum() - diff.diagonal().sum()
    score = (total / (length * (length - 1))) ** (1 / p)
    if reduction in ("elementwise_mean", "mean", "sum", "none", None):
        return score
    raise ValueError(f"Unknown reduction {reduction}")


def spatial_distortion_index(
    preds: Tensor,
    target: Dict[str, Tensor],
    norm_order: int = 1,
    window_size: int = 7,
    reduction: str = "elementwise_mean",
) -> Tensor:
    """D_s: spatial distortion; ``target`` dict holds 'ms', 'pan' (and optionally 'pan_lr')."""
    if not all(k in target for k in ("ms", "pan")):
        raise ValueError(f"Expected `target` to have keys ('ms', 'pan'). Got target: {target.keys()}.")
    ms, pan = target["ms"], target["pan"]
    if preds.ndim != 4:
        raise ValueError(f"Expected `preds` to have BxCxHxW shape. Got preds: {preds.shape}.")
    length = preds.shape[1]

    pan_lr = target.get("pan_lr")
    if pan_lr is None:
        # degrade pan to the MS resolution by average pooling
        ratio = pan.shape[-1] // ms.shape[-1]
        pan_lr = F.avg_pool2d(pan, kernel_size=ratio) if ratio > 1 else pan

    ds = torch.zeros(length, device=preds.device)
    for i in range(length):
        q_hr = _band_uqi(preds[:, i], pan[:, 0], window_size).mean()
        q_lr = _band_uqi(ms[:, i], pan_lr[:, 0], window_size).mean()
        ds[i] = (q_hr - q_lr).abs() ** norm_order
    return (ds.mean()) ** (1 / norm_order)


def quality_with_no_reference(
    preds: Tensor,
    target: Dict[str, Tensor],
    alpha: float = 1.0,
    beta: float = 1.0,
    norm_order: int = 1,
    window_size: int = 7,
    reduction: str = "elementwise_mean",
) -> Tensor:
    """QNR = (1 - D_lambda)^alpha * (1 - D_s)^beta."""
    d_lambda = spectral_distortion_index(preds, target["ms"], p=norm_order, reduction=reduction)
    d_s = spatial_distortion_index(preds, target, norm_order, window_size, reduction)
    return (1 - d_lambda) ** alpha * (1 - d_s) ** beta
