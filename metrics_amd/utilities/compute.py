"""L0 numeric helpers.

Parity: torchmetrics ``utilities/compute.py`` (_safe_divide, _safe_xlogy,
_safe_matmul, _auc_compute, _adjust_weights_safe_divide,
normalize_logits_if_needed).
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from metrics_amd.utilities.data import interp  # noqa: F401  (re-export)


def _safe_matmul(x: Tensor, y: Tensor) -> Tensor:
    """Matmul that upcasts half dtypes to fp32 for the accumulate.

    On MI355X the bf16 MFMA path accumulates in fp32 natively (hipBLASLt), so
    this only changes the *output* dtype to fp32 for numerically-sensitive
    consumers.
    """
    if x.is_floating_point() and x.dtype in (torch.float16, torch.bfloat16):
        return (x.float() @ y.float())
    return x @ y


def _safe_xlogy(x: Tensor, y: Tensor) -> Tensor:
    """x * log(y), defined as 0 where x == 0."""
    res = x * torch.log(y)
    res[x == 0] = 0.0
    return res


def _safe_divide(num: Tensor, denom: Tensor, zero_division: float = 0.0) -> Tensor:
    """num / denom with 0-denominator entries replaced by ``zero_division``."""
    num = num if num.is_floating_point() else num.float()
    denom = denom if denom.is_floating_point() else denom.float()
    zero_division_t = torch.tensor(zero_division, dtype=num.dtype, device=num.device)
    return torch.where(denom != 0, num / denom, zero_division_t)


def _adjust_weights_safe_divide(
    score: Tensor, average: Optional[str], multilabel: bool, tp: Tensor, fp: Tensor, fn: Tensor,
    top_k: int = 1,
) -> Tensor:
    """Apply macro / weighted averaging over per-class scores, skipping empty classes."""
    if average is None or average == "none":
        return score
    if average == "weighted":
        weights = tp + fn
    else:  # macro
        weights = torch.ones_like(score)
        if not multilabel:
            weights[tp + fp + fn == 0 if top_k == 1 else tp + fn == 0] = 0.0
    return _safe_divide(weights * score, weights.sum(-1, keepdim=True)).sum(-1)


def _auc_compute_without_check(x: Tensor, y: Tensor, direction: float, axis: int = -1) -> Tensor:
    """Trapezoidal area under (x, y); ``direction`` flips sign for decreasing x."""
    with torch.no_grad():
        auc_score = torch.trapz(y, x, dim=axis) * direction
    return auc_score


def _auc_compute(x: Tensor, y: Tensor, reorder: bool = False) -> Tensor:
    with torch.no_grad():
        if reorder:
            x, x_idx = torch.sort(x, stable=True)
            y = y[x_idx]
        dx = x[1:] - x[:-1]
        if (dx < 0).any():
            if (dx <= 0).all():
                direction = -1.0
            else:
                raise ValueError(
                    "The `x` tensor is neither increasing or decreasing. Try setting reorder=True."
                )
        else:
            direction = 1.0
        return _auc_compute_without_check(x, y, direction)


def auc(x: Tensor, y: Tensor, reorder: bool = False) -> Tensor:
    """Area under the curve y = f(x) using the trapezoidal rule."""
    if x.ndim > 1 or y.ndim > 1:
        raise ValueError(f"Expected both x and y to be 1d, got {x.ndim}d and {y.ndim}d")
    if x.numel() != y.numel():
        raise ValueError("Expected the same number of elements in x and y")
    return _auc_compute(x, y, reorder=reorder)


def normalize_logits_if_needed(tensor: Tensor, normalization: str) -> Tensor:
    """Apply sigmoid/softmax iff values fall outside [0, 1].

    Single-pass check; we control the device and accept the device sync the
    ``.any()`` implies only on CPU — on GPU the branchless torch.where form is
    used to avoid a host sync (the hot update kernels bypass this entirely and
    normalize in-kernel).
    """
    assert normalization in ("sigmoid", "softmax")
    if tensor.device.type == "cpu":
        if not torch.all((tensor >= 0) * (tensor <= 1)):
            tensor = tensor.sigmoid() if normalization == "sigmoid" else tensor.softmax(dim=-1)
        return tensor
    # branchless on device: no host<->device sync
    condition = ((tensor < 0) | (tensor > 1)).any()
    return torch.where(
        condition,
        torch.sigmoid(tensor) if normalization == "sigmoid" else torch.softmax(tensor, dim=-1),
        tensor,
    )
