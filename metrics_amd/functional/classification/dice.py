"""Dice coefficient functional (legacy-style API).

Parity: reference functional/classification/dice.py:68 (the deprecated
``dice``); matches our modular ``metrics_amd.classification.Dice``.
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from metrics_amd.utilities.compute import _safe_divide
from metrics_amd.functional.classification.stat_scores import (
    _multiclass_stat_scores_arg_validation,
    _multiclass_stat_scores_format,
    _multiclass_stat_scores_tensor_validation,
    _multiclass_stat_scores_update,
)


def dice(
    preds: Tensor,
    target: Tensor,
    zero_division: int = 0,
    average: Optional[str] = "micro",
    mdmc_average: Optional[str] = "global",
    threshold: float = 0.5,
    top_k: Optional[int] = None,
    num_classes: Optional[int] = None,
    multiclass: Optional[bool] = None,
    ignore_index: Optional[int] = None,
) -> Tensor:
    """Dice = 2*TP / (2*TP + FP + FN) with micro/macro/weighted/samples-free averaging."""
    if mdmc_average not in ("global", None):
        raise NotImplementedError(
            "dice(mdmc_average='samplewise') belongs to the reference's deprecated legacy input machinery"
            " (removed in its v1.7); use functional.segmentation.dice_score."
        )
    if multiclass is not None:
        raise NotImplementedError(
            "dice(multiclass=...) input coercion belongs to the reference's deprecated legacy machinery"
            " (removed in its v1.7); pass explicitly shaped inputs instead."
        )
    if average not in ("micro", "macro", "weighted", "none", None):
        raise ValueError(f"The `average` has to be one of 'micro'/'macro'/'weighted'/'none', got {average}.")
    if num_classes is None:
        # legacy API: infer the class count from the inputs (reference dice.py
        # routes through the legacy input-format classifier)
        if preds.is_floating_point() and preds.ndim == target.ndim + 1:
            num_classes = preds.shape[1]
        else:
            num_classes = int(torch.max(torch.stack([preds.max(), target.max()])).item()) + 1
    _multiclass_stat_scores_arg_validation(num_classes, top_k or 1, average or "micro", "global", ignore_index)
    _multiclass_stat_scores_tensor_validation(preds, target, num_classes, "global", ignore_index)
    preds_f, target_f = _multiclass_stat_scores_format(preds, target, top_k or 1)
    tp, fp, tn, fn = _multiclass_stat_scores_update(
        preds_f, target_f, num_classes, top_k or 1, average or "micro", "global", ignore_index
    )
    if average == "micro":
        tp, fp, fn = tp.sum(), fp.sum(), fn.sum()
        return _safe_divide(2 * tp, 2 * tp + fp + fn, zero_division)
    score = _safe_divide(2 * tp, 2 * tp + fp + fn, zero_division)
    if average == "macro":
        return score.float().mean()
    if average == "weighted":
        w = tp + fn
        return (score * _safe_divide(w, w.sum())).sum()
    return score
