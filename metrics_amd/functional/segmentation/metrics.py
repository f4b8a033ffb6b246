"""Segmentation metrics (functional).

Parity: torchmetrics ``functional/segmentation/{mean_iou,dice,generalized_dice,
hausdorff_distance}.py``.
"""
from __future__ import annotations

from typing import Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.compute import _safe_divide
from metrics_amd.functional.segmentation.utils import _ignore_background, edge_mask, surface_distance


def _check_input_format(input_format: str) -> None:
    if input_format not in ("one-hot", "index", "mixed"):
        raise ValueError(f"Expected argument `input_format` to be one of 'one-hot', 'index', 'mixed', but got {input_format}")


def _format_inputs(preds: Tensor, target: Tensor, num_classes: int, input_format: str) -> Tuple[Tensor, Tensor]:
    """Convert to one-hot (N, C, ...) layout."""
    if input_format == "index":
        preds = torch.nn.functional.one_hot(preds.long(), num_classes=num_classes).movedim(-1, 1)
        target = torch.nn.functional.one_hot(target.long(), num_classes=num_classes).movedim(-1, 1)
    elif input_format == "mixed":
        if preds.ndim == target.ndim + 1:
            target = torch.nn.functional.one_hot(target.long(), num_classes=num_classes).movedim(-1, 1)
        elif target.ndim == preds.ndim + 1:
            preds = torch.nn.functional.one_hot(preds.long(), num_classes=num_classes).movedim(-1, 1)
    return preds, target


def _mean_iou_update(
    preds: Tensor,
    target: Tensor,
    num_classes: Optional[int] = None,
    include_background: bool = False,
    input_format: str = "one-hot",
) -> Tuple[Tensor, Tensor]:
    """Per-sample per-class (intersection, union)."""
    _check_input_format(input_format)
    if input_format in ("index", "mixed") and num_classes is None:
        raise ValueError("Argument `num_classes` must be provided when `input_format` is 'index' or 'mixed'")
    preds, target = _format_inputs(preds, target, num_classes, input_format)
    if not include_background:
        preds, target = _ignore_background(preds, target)

    reduce_axis = list(range(2, preds.ndim))
    intersection = torch.sum(preds * target, dim=reduce_axis)
    target_sum = torch.sum(target, dim=reduce_axis)
    pred_sum = torch.sum(preds, dim=reduce_axis)
    union = target_sum + pred_sum - intersection
    return intersection, union


def _mean_iou_compute(intersection: Tensor, union: Tensor, zero_division: Union[float, str] = "warn") -> Tensor:
    """Per-sample per-class IoU."""
    valid = union > 0
    iou = _safe_divide(intersection, union, zero_division=0.0 if zero_division in ("warn", "nan") else zero_division)
    if zero_division == "nan":
        iou = torch.where(valid, iou, torch.tensor(float("nan"), device=iou.device))
    return iou


def mean_iou(
    preds: Tensor,
    target: Tensor,
    num_classes: Optional[int] = None,
    include_background: bool = True,
    per_class: bool = False,
    input_format: str = "one-hot",
) -> Tensor:
    """Mean intersection over union for semantic segmentation."""
    intersection, union = _mean_iou_update(preds, target, num_classes, include_background, input_format)
    iou = _mean_iou_compute(intersection, union, zero_division="nan")
    return iou.nanmean(0) if per_class else iou.nanmean(-1)


def _dice_score_update(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    include_background: bool = True,
    input_format: str = "one-hot",
) -> Tuple[Tensor, Tensor, Tensor]:
    _check_input_format(input_format)
    preds, target = _format_inputs(preds, target, num_classes, input_format)
    if not include_background:
        preds, target = _ignore_background(preds, target)

    reduce_axis = list(range(2, preds.ndim))
    intersection = torch.sum(preds * target, dim=reduce_axis)
    target_sum = torch.sum(target, dim=reduce_axis)
    pred_sum = torch.sum(preds, dim=reduce_axis)

    numerator = 2 * intersection
    denominator = pred_sum + target_sum
    support = target_sum
    return numerator, denominator, support


def _dice_score_compute(
    numerator: Tensor, denominator: Tensor, average: Optional[str] = "micro", support: Optional[Tensor] = None
) -> Tensor:
    """Reduce per-sample per-class dice parts."""
    if average == "micro":
        numerator = numerator.sum(-1)
        denominator = denominator.sum(-1)
        dice = _safe_divide(numerator, denominator, zero_division=float("nan"))
    else:
        dice = _safe_divide(numerator, denominator, zero_division=float("nan"))
        if average == "macro":
            dice = dice.nanmean(-1)
        elif average == "weighted" and support is not None:
            weights = _safe_divide(support, support.sum(-1, keepdim=True))
            dice = (dice * weights).nansum(-1)
    return dice


def dice_score(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    include_background: bool = True,
    average: Optional[str] = "micro",
    input_format: str = "one-hot",
) -> Tensor:
    """Dice score for semantic segmentation (per-sample, then user averages)."""
    numerator, denominator, support = _dice_score_update(preds, target, num_classes, include_background, input_format)
    return _dice_score_compute(numerator, denominator, average, support)


def generalized_dice_score(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    include_background: bool = True,
    per_class: bool = False,
    weight_type: str = "square",
    input_format: str = "one-hot",
) -> Tensor:
    """Generalized dice score with square/simple/linear class weighting."""
    _check_input_format(input_format)
    if weight_type not in ("square", "simple", "linear"):
        raise ValueError(f"Expected argument `weight_type` to be one of 'square', 'simple', 'linear', but got {weight_type}")
    preds, target = _format_inputs(preds, target, num_classes, input_format)
    if not include_background:
        preds, target = _ignore_background(preds, target)

    reduce_axis = list(range(2, preds.ndim))
    intersection = torch.sum(preds * target, dim=reduce_axis)
    target_sum = torch.sum(target, dim=reduce_axis)
    pred_sum = torch.sum(preds, dim=reduce_axis)

    if weight_type == "simple":
        weights = 1.0 / target_sum
    elif weight_type == "linear":
        weights = torch.ones_like(target_sum)
    else:  # square
        weights = 1.0 / (target_sum**2)

    w_inf = torch.isinf(weights)
    weights[w_inf] = 0.0
    weights = torch.where(w_inf, weights.max(dim=1, keepdim=True).values.expand_as(weights), weights)

    numerator = 2.0 * intersection * weights
    denominator = (pred_sum + target_sum) * weights
    if per_class:
        return _safe_divide(numerator, denominator)
    return _safe_divide(numerator.sum(-1), denominator.sum(-1))


def hausdorff_distance(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    include_background: bool = False,
    distance_metric: str = "euclidean",
    spacing=None,
    directed: bool = False,
    input_format: str = "one-hot",
) -> Tensor:
    """(Symmetric) Hausdorff distance between segmentation boundaries, per class averaged over batch."""
    _check_input_format(input_format)
    preds, target = _format_inputs(preds, target, num_classes, input_format)
    if not include_background:
        preds, target = _ignore_background(preds, target)

    n, c = preds.shape[0], preds.shape[1]
    out = torch.zeros(n, c, device=preds.device)
    for i in range(n):
        for j in range(c):
            pe = edge_mask(preds[i, j])
            te = edge_mask(target[i, j])
            d_pt = surface_distance(pe, te, distance_metric, spacing)
            if directed:
                out[i, j] = d_pt.max()
            else:
                d_tp = surface_distance(te, pe, distance_metric, spacing)
                out[i, j] = torch.maximum(d_pt.max(), d_tp.max())
    return out.mean(0)
