"""Group fairness metrics. Parity: torchmetrics ``functional/classification/group_fairness.py``."""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.utilities.compute import _safe_divide
from metrics_amd.functional.classification.stat_scores import (
    _binary_stat_scores_arg_validation,
    _binary_stat_scores_format,
    _binary_stat_scores_tensor_validation,
    _binary_stat_scores_update,
)


def _groups_validation(groups: Tensor, num_groups: int) -> None:
    if torch.max(groups) > num_groups - 1:
        raise ValueError(
            f"The largest number in the groups tensor is {torch.max(groups)}, which is larger than the specified"
            f" number of groups {num_groups}. The group identifiers should be ``0, 1, ..., num_groups - 1``."
        )
    if groups.dtype not in (torch.int16, torch.int32, torch.int64, torch.uint8, torch.int8):
        raise ValueError(f"Expected dtype of argument groups to be int, but got {groups.dtype}.")


def _groups_format(groups: Tensor) -> Tensor:
    return groups.reshape(groups.shape[0], -1)


def _binary_groups_stat_scores(
    preds: Tensor,
    target: Tensor,
    groups: Tensor,
    num_groups: int,
    threshold: float = 0.5,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> List[Tuple[Tensor, Tensor, Tensor, Tensor]]:
    """Per-group (tp, fp, tn, fn)."""
    if validate_args:
        _binary_stat_scores_arg_validation(threshold, "global", ignore_index)
        _binary_stat_scores_tensor_validation(preds, target, "global", ignore_index)
        _groups_validation(groups, num_groups)

    preds, target = _binary_stat_scores_format(preds, target, threshold, ignore_index)
    groups = _groups_format(groups)

    indexes, indices = torch.sort(groups.squeeze(1))
    preds = preds.squeeze(1)[indices]
    target = target.squeeze(1)[indices]

    split_sizes = torch.bincount(indexes, minlength=num_groups).tolist()
    group_preds = list(torch.split(preds, split_sizes, dim=0))
    group_target = list(torch.split(target, split_sizes, dim=0))
    # _binary_stat_scores_update expects the (N, flattened...) 2D layout
    return [_binary_stat_scores_update(p.unsqueeze(-1), t.unsqueeze(-1)) for p, t in zip(group_preds, group_target)]


def binary_groups_stat_rates(
    preds: Tensor,
    target: Tensor,
    groups: Tensor,
    num_groups: int,
    threshold: float = 0.5,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Dict[str, Tensor]:
    """Per-group [tp_rate, fp_rate, tn_rate, fn_rate]."""
    stats = _binary_groups_stat_scores(preds, target, groups, num_groups, threshold, ignore_index, validate_args)
    out = {}
    for group, (tp, fp, tn, fn) in enumerate(stats):
        total = tp + fp + tn + fn
        out[f"group_{group}"] = torch.stack([tp, fp, tn, fn]) / total
    return out


def _compute_binary_demographic_parity(tp: Tensor, fp: Tensor, tn: Tensor, fn: Tensor) -> Tensor:
    """Positive prediction rate per group."""
    return _safe_divide(tp + fp, tp + fp + tn + fn)


def _compute_binary_equal_opportunity(tp: Tensor, fp: Tensor, tn: Tensor, fn: Tensor) -> Tensor:
    """True positive rate per group."""
    return _safe_divide(tp, tp + fn)


def binary_fairness(
    preds: Tensor,
    target: Tensor,
    groups: Tensor,
    task: str = "all",
    threshold: float = 0.5,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Dict[str, Tensor]:
    """Demographic parity and/or equal opportunity ratios between groups."""
    if task not in ("demographic_parity", "equal_opportunity", "all"):
        raise ValueError(
            f"Expected argument `task` to either be 'demographic_parity', 'equal_opportunity' or 'all' but got {task}."
        )
    num_groups = int(torch.max(groups)) + 1
    if task == "demographic_parity":
        # target is ignored for demographic parity
        target = torch.zeros_like(preds, dtype=torch.long)
        validate_args = False
    stats = _binary_groups_stat_scores(preds, target, groups, num_groups, threshold, ignore_index, validate_args)
    tps = torch.stack([s[0] for s in stats])
    fps = torch.stack([s[1] for s in stats])
    tns = torch.stack([s[2] for s in stats])
    fns = torch.stack([s[3] for s in stats])

    out: Dict[str, Tensor] = {}
    if task in ("demographic_parity", "all"):
        rates = _compute_binary_demographic_parity(tps, fps, tns, fns)
        min_g = int(torch.argmin(rates))
        max_g = int(torch.argmax(rates))
        out[f"DP_{min_g}_{max_g}"] = _safe_divide(rates[min_g], rates[max_g])
    if task in ("equal_opportunity", "all"):
        rates = _compute_binary_equal_opportunity(tps, fps, tns, fns)
        min_g = int(torch.argmin(rates))
        max_g = int(torch.argmax(rates))
        out[f"EO_{min_g}_{max_g}"] = _safe_divide(rates[min_g], rates[max_g])
    return out


def demographic_parity(
    preds: Tensor,
    groups: Tensor,
    threshold: float = 0.5,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Dict[str, Tensor]:
    """Demographic parity: min/max positivity-rate ratio between groups.

    Parity: reference functional/classification/group_fairness.py:177.
    """
    return binary_fairness(preds, torch.zeros_like(preds, dtype=torch.long), groups,
                           "demographic_parity", threshold, ignore_index, validate_args)


def equal_opportunity(
    preds: Tensor,
    target: Tensor,
    groups: Tensor,
    threshold: float = 0.5,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Dict[str, Tensor]:
    """Equal opportunity: min/max true-positive-rate ratio between groups.

    Parity: reference functional/classification/group_fairness.py:258.
    """
    return binary_fairness(preds, target, groups, "equal_opportunity", threshold, ignore_index, validate_args)
