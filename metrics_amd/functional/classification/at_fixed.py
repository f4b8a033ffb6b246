"""At-fixed-X curve metrics: recall@precision, precision@recall,
sensitivity@specificity, specificity@sensitivity.

Parity: torchmetrics ``functional/classification/{recall_fixed_precision,
precision_fixed_recall,sensitivity_specificity,specificity_sensitivity}.py``.
All reuse the PR-curve / ROC state machinery (HIP bucketized histograms on GPU).
"""
from __future__ import annotations

from typing import List, Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.precision_recall_curve import (
    _binary_precision_recall_curve_arg_validation,
    _binary_precision_recall_curve_compute,
    _binary_precision_recall_curve_format,
    _binary_precision_recall_curve_tensor_validation,
    _binary_precision_recall_curve_update,
    _multiclass_precision_recall_curve_arg_validation,
    _multiclass_precision_recall_curve_compute,
    _multiclass_precision_recall_curve_format,
    _multiclass_precision_recall_curve_tensor_validation,
    _multiclass_precision_recall_curve_update,
    _multilabel_precision_recall_curve_arg_validation,
    _multilabel_precision_recall_curve_compute,
    _multilabel_precision_recall_curve_format,
    _multilabel_precision_recall_curve_tensor_validation,
    _multilabel_precision_recall_curve_update,
)
from metrics_amd.functional.classification.roc import (
    _binary_roc_compute,
    _multiclass_roc_compute,
    _multilabel_roc_compute,
)


def _lexargmax(x: Tensor) -> Tensor:
    """Index of the lexicographic maximum row (reference recall_fixed_precision.py:40)."""
    idx = None
    for k in range(x.shape[1]):
        col = x[idx, k] if idx is not None else x[:, k]
        z = torch.where(col == col.max())[0]
        idx = z if idx is None else idx[z]
        if len(idx) < 2:
            break
    if idx is None:
        raise ValueError("Failed to extract index")
    return idx


def _recall_at_precision(precision: Tensor, recall: Tensor, thresholds: Tensor, min_precision: float):
    """Reference recall_fixed_precision.py:58 — lexicographic max over
    (recall, precision, threshold) among rows with precision >= min."""
    max_recall = torch.tensor(0.0, device=recall.device, dtype=recall.dtype)
    best_threshold = torch.tensor(0)
    zipped_len = min(t.shape[0] for t in (recall, precision, thresholds))
    zipped = torch.vstack((recall[:zipped_len], precision[:zipped_len], thresholds[:zipped_len])).T
    zipped_masked = zipped[zipped[:, 1] >= min_precision]
    if zipped_masked.shape[0] > 0:
        idx = _lexargmax(zipped_masked)[0]
        max_recall, _, best_threshold = zipped_masked[idx]
    if max_recall == 0.0:
        best_threshold = torch.tensor(1e6, device=thresholds.device, dtype=thresholds.dtype)
    return max_recall, best_threshold


def _precision_at_recall(precision: Tensor, recall: Tensor, thresholds: Tensor, min_recall: float):
    """Reference precision_fixed_recall.py:42 — python tuple-max over (p, r, t)."""
    try:
        max_precision, _, best_threshold = max(
            (p, r, t) for p, r, t in zip(precision, recall, thresholds) if r >= min_recall
        )
    except ValueError:
        max_precision = torch.tensor(0.0, device=precision.device, dtype=precision.dtype)
        best_threshold = torch.tensor(0)
    if max_precision == 0.0:
        best_threshold = torch.tensor(1e6, device=thresholds.device, dtype=thresholds.dtype)
    return max_precision, best_threshold


def binary_recall_at_fixed_precision(
    preds: Tensor,
    target: Tensor,
    min_precision: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """(max recall with precision >= min_precision, its threshold)."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
        if not isinstance(min_precision, float) or not (0 <= min_precision <= 1):
            raise ValueError(f"Expected argument `min_precision` to be a float in the [0,1] range, but got {min_precision}")
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _binary_precision_recall_curve_format(preds, target, thresholds, ignore_index, remove_ignored=remove_ignored)
    state = _binary_precision_recall_curve_update(p_f, t_f, thr, ignore_index if not remove_ignored else None)
    precision, recall, thresholds_out = _binary_precision_recall_curve_compute(state, thr)
    return _recall_at_precision(precision, recall, thresholds_out, min_precision)


def multiclass_recall_at_fixed_precision(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    min_precision: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-class (recall, threshold) at fixed precision."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        p_f, t_f, num_classes, thr, None, ignore_index if not remove_ignored else None
    )
    precision, recall, thresholds_out = _multiclass_precision_recall_curve_compute(state, num_classes, thr, None)
    if isinstance(precision, Tensor):
        res = [
            _recall_at_precision(precision[i], recall[i], thresholds_out, min_precision) for i in range(num_classes)
        ]
    else:
        res = [
            _recall_at_precision(precision[i], recall[i], thresholds_out[i], min_precision) for i in range(num_classes)
        ]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def multilabel_recall_at_fixed_precision(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    min_precision: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-label (recall, threshold) at fixed precision."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    p_f, t_f, thr = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(p_f, t_f, num_labels, thr, ignore_index if not remove_ignored else None)
    precision, recall, thresholds_out = _multilabel_precision_recall_curve_compute(state, num_labels, thr, ignore_index)
    if isinstance(precision, Tensor):
        res = [_recall_at_precision(precision[i], recall[i], thresholds_out, min_precision) for i in range(num_labels)]
    else:
        res = [_recall_at_precision(precision[i], recall[i], thresholds_out[i], min_precision) for i in range(num_labels)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def binary_precision_at_fixed_recall(
    preds: Tensor,
    target: Tensor,
    min_recall: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """(max precision with recall >= min_recall, its threshold)."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _binary_precision_recall_curve_format(preds, target, thresholds, ignore_index, remove_ignored=remove_ignored)
    state = _binary_precision_recall_curve_update(p_f, t_f, thr, ignore_index if not remove_ignored else None)
    precision, recall, thresholds_out = _binary_precision_recall_curve_compute(state, thr)
    return _precision_at_recall(precision, recall, thresholds_out, min_recall)


def multiclass_precision_at_fixed_recall(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    min_recall: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-class (precision, threshold) at fixed recall."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        p_f, t_f, num_classes, thr, None, ignore_index if not remove_ignored else None
    )
    precision, recall, thresholds_out = _multiclass_precision_recall_curve_compute(state, num_classes, thr, None)
    if isinstance(precision, Tensor):
        res = [_precision_at_recall(precision[i], recall[i], thresholds_out, min_recall) for i in range(num_classes)]
    else:
        res = [_precision_at_recall(precision[i], recall[i], thresholds_out[i], min_recall) for i in range(num_classes)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def multilabel_precision_at_fixed_recall(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    min_recall: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-label (precision, threshold) at fixed recall."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    p_f, t_f, thr = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(p_f, t_f, num_labels, thr, ignore_index if not remove_ignored else None)
    precision, recall, thresholds_out = _multilabel_precision_recall_curve_compute(state, num_labels, thr, ignore_index)
    if isinstance(precision, Tensor):
        res = [_precision_at_recall(precision[i], recall[i], thresholds_out, min_recall) for i in range(num_labels)]
    else:
        res = [_precision_at_recall(precision[i], recall[i], thresholds_out[i], min_recall) for i in range(num_labels)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


# ---------------------------------------------------------------- ROC variants
def _sens_at_spec(fpr: Tensor, tpr: Tensor, thresholds: Tensor, min_specificity: float):
    """Reference sensitivity_specificity.py:47 — argmax sensitivity where spec >= min."""
    specificity = 1 - fpr
    indices = specificity >= min_specificity
    if not indices.any():
        return (torch.tensor(0.0, device=tpr.device, dtype=tpr.dtype),
                torch.tensor(1e6, device=thresholds.device, dtype=thresholds.dtype))
    sens, thr = tpr[indices], thresholds[indices]
    idx = torch.argmax(sens)
    return sens[idx], thr[idx]


def _spec_at_sens(fpr: Tensor, tpr: Tensor, thresholds: Tensor, min_sensitivity: float):
    """Reference specificity_sensitivity.py:48 — argmax specificity where sens >= min."""
    specificity = 1 - fpr
    indices = tpr >= min_sensitivity
    if not indices.any():
        return (torch.tensor(0.0, device=specificity.device, dtype=specificity.dtype),
                torch.tensor(1e6, device=thresholds.device, dtype=thresholds.dtype))
    spec, thr = specificity[indices], thresholds[indices]
    idx = torch.argmax(spec)
    return spec[idx], thr[idx]


def binary_sensitivity_at_specificity(
    preds: Tensor,
    target: Tensor,
    min_specificity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """(max sensitivity with specificity >= min_specificity, threshold)."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _binary_precision_recall_curve_format(preds, target, thresholds, ignore_index, remove_ignored=remove_ignored)
    state = _binary_precision_recall_curve_update(p_f, t_f, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, thresholds_out = _binary_roc_compute(state, thr)
    return _sens_at_spec(fpr, tpr, thresholds_out, min_specificity)


def multiclass_sensitivity_at_specificity(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    min_specificity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-class (sensitivity, threshold) at fixed specificity."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        p_f, t_f, num_classes, thr, None, ignore_index if not remove_ignored else None
    )
    fpr, tpr, thresholds_out = _multiclass_roc_compute(state, num_classes, thr)
    if isinstance(fpr, Tensor):
        res = [_sens_at_spec(fpr[i], tpr[i], thresholds_out, min_specificity) for i in range(num_classes)]
    else:
        res = [_sens_at_spec(fpr[i], tpr[i], thresholds_out[i], min_specificity) for i in range(num_classes)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def multilabel_sensitivity_at_specificity(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    min_specificity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-label (sensitivity, threshold) at fixed specificity."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    p_f, t_f, thr = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(p_f, t_f, num_labels, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, thresholds_out = _multilabel_roc_compute(state, num_labels, thr, ignore_index)
    if isinstance(fpr, Tensor):
        res = [_sens_at_spec(fpr[i], tpr[i], thresholds_out, min_specificity) for i in range(num_labels)]
    else:
        res = [_sens_at_spec(fpr[i], tpr[i], thresholds_out[i], min_specificity) for i in range(num_labels)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def binary_specificity_at_sensitivity(
    preds: Tensor,
    target: Tensor,
    min_sensitivity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """(max specificity with sensitivity >= min_sensitivity, threshold)."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _binary_precision_recall_curve_format(preds, target, thresholds, ignore_index, remove_ignored=remove_ignored)
    state = _binary_precision_recall_curve_update(p_f, t_f, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, thresholds_out = _binary_roc_compute(state, thr)
    return _spec_at_sens(fpr, tpr, thresholds_out, min_sensitivity)


def multiclass_specificity_at_sensitivity(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    min_sensitivity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-class (specificity, threshold) at fixed sensitivity."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        p_f, t_f, num_classes, thr, None, ignore_index if not remove_ignored else None
    )
    fpr, tpr, thresholds_out = _multiclass_roc_compute(state, num_classes, thr)
    if isinstance(fpr, Tensor):
        res = [_spec_at_sens(fpr[i], tpr[i], thresholds_out, min_sensitivity) for i in range(num_classes)]
    else:
        res = [_spec_at_sens(fpr[i], tpr[i], thresholds_out[i], min_sensitivity) for i in range(num_classes)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


def multilabel_specificity_at_sensitivity(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    min_sensitivity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor]:
    """Per-label (specificity, threshold) at fixed sensitivity."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    p_f, t_f, thr = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(p_f, t_f, num_labels, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, thresholds_out = _multilabel_roc_compute(state, num_labels, thr, ignore_index)
    if isinstance(fpr, Tensor):
        res = [_spec_at_sens(fpr[i], tpr[i], thresholds_out, min_sensitivity) for i in range(num_labels)]
    else:
        res = [_spec_at_sens(fpr[i], tpr[i], thresholds_out[i], min_sensitivity) for i in range(num_labels)]
    return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


# -------------------------------------------------------------------- LogAUC
def _logauc_compute(fpr: Tensor, tpr: Tensor, fpr_range: Tuple[float, float]) -> Tensor:
    """Area under the ROC in log10(fpr) space over ``fpr_range``, normalized.

    Reference functional/classification/logauc.py:35 (_binary_logauc_compute):
    interpolated bounds are appended and fpr/tpr sorted INDEPENDENTLY (both
    are monotone along the curve), then the log-space trapezoid is taken
    between the last occurrences of the two bounds.
    """
    from metrics_amd.utilities.data import interp
    from metrics_amd.utilities.prints import rank_zero_warn

    fpr_range_t = torch.tensor(fpr_range).to(fpr.device)
    if fpr.numel() < 2 or tpr.numel() < 2:
        rank_zero_warn(
            "At least two values on for the fpr and tpr are required to compute the log AUC. Returns 0 score."
        )
        return torch.tensor(0.0, device=fpr.device)

    tpr = torch.cat([tpr, interp(fpr_range_t, fpr, tpr)]).sort().values
    fpr = torch.cat([fpr, fpr_range_t]).sort().values

    log_fpr = torch.log10(fpr)
    bounds = torch.log10(torch.tensor(fpr_range))

    lower_bound_idx = torch.where(log_fpr == bounds[0])[0][-1]
    upper_bound_idx = torch.where(log_fpr == bounds[1])[0][-1]

    trimmed_log_fpr = log_fpr[lower_bound_idx : upper_bound_idx + 1]
    trimmed_tpr = tpr[lower_bound_idx : upper_bound_idx + 1]

    from metrics_amd.utilities.compute import _auc_compute_without_check

    return _auc_compute_without_check(trimmed_log_fpr, trimmed_tpr, 1.0) / (bounds[1] - bounds[0])


def binary_logauc(
    preds: Tensor,
    target: Tensor,
    fpr_range: Tuple[float, float] = (0.001, 0.1),
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Log-scaled AUC over an fpr range (virtual-screening style)."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
        if not isinstance(fpr_range, (tuple, list)) or len(fpr_range) != 2 or fpr_range[0] >= fpr_range[1]:
            raise ValueError(f"The `fpr_range` should be a tuple of two floats (low, high), but got {fpr_range}")
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _binary_precision_recall_curve_format(preds, target, thresholds, ignore_index, remove_ignored=remove_ignored)
    state = _binary_precision_recall_curve_update(p_f, t_f, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, _ = _binary_roc_compute(state, thr)
    return _logauc_compute(fpr, tpr, tuple(fpr_range))


def multiclass_logauc(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    fpr_range: Tuple[float, float] = (0.001, 0.1),
    average: Optional[str] = "macro",
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Per-class (or macro) log AUC."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    p_f, t_f, thr = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        p_f, t_f, num_classes, thr, None, ignore_index if not remove_ignored else None
    )
    fpr, tpr, _ = _multiclass_roc_compute(state, num_classes, thr)
    scores = torch.stack([_logauc_compute(fpr[i], tpr[i], tuple(fpr_range)) for i in range(num_classes)])
    if average == "macro":
        return scores.mean()
    return scores


def multilabel_logauc(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    fpr_range: Tuple[float, float] = (0.001, 0.1),
    average: Optional[str] = "macro",
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Per-label (or macro) log AUC."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    p_f, t_f, thr = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(p_f, t_f, num_labels, thr, ignore_index if not remove_ignored else None)
    fpr, tpr, _ = _multilabel_roc_compute(state, num_labels, thr, ignore_index)
    scores = torch.stack([_logauc_compute(fpr[i], tpr[i], tuple(fpr_range)) for i in range(num_labels)])
    if average == "macro":
        return scores.mean()
    return scores


def recall_at_fixed_precision(
    preds: Tensor,
    target: Tensor,
    task: str,
    min_precision: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tuple[Tensor, Tensor]]:
    """Task-dispatching recall-at-fixed-precision (reference functional/classification/recall_fixed_precision.py:401)."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_recall_at_fixed_precision(preds, target, min_precision, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_recall_at_fixed_precision(
            preds, target, num_classes, min_precision, thresholds, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_recall_at_fixed_precision(
            preds, target, num_labels, min_precision, thresholds, ignore_index, validate_args
        )
    raise ValueError(f"Not handled value: {task}")


def precision_at_fixed_recall(
    preds: Tensor,
    target: Tensor,
    task: str,
    min_recall: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tuple[Tensor, Tensor]]:
    """Task-dispatching precision-at-fixed-recall (reference functional/classification/precision_fixed_recall.py)."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_precision_at_fixed_recall(preds, target, min_recall, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_precision_at_fixed_recall(
            preds, target, num_classes, min_recall, thresholds, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_precision_at_fixed_recall(
            preds, target, num_labels, min_recall, thresholds, ignore_index, validate_args
        )
    raise ValueError(f"Not handled value: {task}")


def sensitivity_at_specificity(
    preds: Tensor,
    target: Tensor,
    task: str,
    min_specificity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tuple[Tensor, Tensor]]:
    """Task-dispatching sensitivity-at-specificity (reference functional/classification/sensitivity_specificity.py)."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_sensitivity_at_specificity(preds, target, min_specificity, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_sensitivity_at_specificity(
            preds, target, num_classes, min_specificity, thresholds, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_sensitivity_at_specificity(
            preds, target, num_labels, min_specificity, thresholds, ignore_index, validate_args
        )
    raise ValueError(f"Not handled value: {task}")


def specificity_at_sensitivity(
    preds: Tensor,
    target: Tensor,
    task: str,
    min_sensitivity: float,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tuple[Tensor, Tensor]]:
    """Task-dispatching specificity-at-sensitivity (reference functional/classification/specificity_sensitivity.py)."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_specificity_at_sensitivity(preds, target, min_sensitivity, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_specificity_at_sensitivity(
            preds, target, num_classes, min_sensitivity, thresholds, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_specificity_at_sensitivity(
            preds, target, num_labels, min_sensitivity, thresholds, ignore_index, validate_args
        )
    raise ValueError(f"Not handled value: {task}")


def logauc(
    preds: Tensor,
    target: Tensor,
    task: str,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    fpr_range: Tuple[float, float] = (0.001, 0.1),
    average: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tensor]:
    """Task-dispatching LogAUC (reference functional/classification/logauc.py:323)."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_logauc(preds, target, fpr_range, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_logauc(preds, target, num_classes, fpr_range, average or "macro", thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_logauc(preds, target, num_labels, fpr_range, average or "macro", thresholds, ignore_index, validate_args)
    raise ValueError(f"Not handled value: {task}")
