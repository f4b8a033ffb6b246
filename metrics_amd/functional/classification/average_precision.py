"""Average precision. Parity: torchmetrics ``functional/classification/average_precision.py``."""
from __future__ import annotations

from typing import List, Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.compute import _safe_divide
from metrics_amd.utilities.data import _bincount
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.utilities.prints import rank_zero_warn
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


def _reduce_average_precision(
    precision: Union[Tensor, List[Tensor]],
    recall: Union[Tensor, List[Tensor]],
    average: Optional[str] = "macro",
    weights: Optional[Tensor] = None,
) -> Tensor:
    """AP = -sum((recall[1:] - recall[:-1]) * precision[:-1]) per curve, then reduce."""
    if isinstance(precision, Tensor) and isinstance(recall, Tensor):
        res = -torch.sum((recall[:, 1:] - recall[:, :-1]) * precision[:, :-1], 1)
    else:
        res = torch.stack([-torch.sum((r[1:] - r[:-1]) * p[:-1]) for p, r in zip(precision, recall)])
    if average is None or average == "none":
        return res
    if torch.isnan(res).any():
        rank_zero_warn(
            "Average precision score for one or more classes was `nan`. Ignoring these classes in average",
            UserWarning,
        )
    idx = ~torch.isnan(res)
    if average == "macro":
        return res[idx].mean()
    if average == "weighted" and weights is not None:
        weights = _safe_divide(weights[idx], weights[idx].sum())
        return (res[idx] * weights).sum()
    raise ValueError("Received an incompatible combinations of inputs to make reduction.")


def _binary_average_precision_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    thresholds: Optional[Tensor],
    pos_label: int = 1,
) -> Tensor:
    if isinstance(state, Tensor) and thresholds is not None and state.is_cuda and state.ndim == 3:
        from metrics_amd.ops import _hip

        if _hip.hip_available():
            res, _ = _hip.curve_auc_from_confmat(state.unsqueeze(1), mode=1)
            return res.reshape(())
    precision, recall, _ = _binary_precision_recall_curve_compute(state, thresholds, pos_label)
    return -torch.sum((recall[1:] - recall[:-1]) * precision[:-1])


def binary_average_precision(
    preds: Tensor,
    target: Tensor,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Average precision for binary tasks."""
    if validate_args:
        _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        _binary_precision_recall_curve_tensor_validation(preds, target, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    preds_f, target_f, thresholds_t = _binary_precision_recall_curve_format(
        preds, target, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _binary_precision_recall_curve_update(
        preds_f, target_f, thresholds_t, ignore_index if not remove_ignored else None
    )
    return _binary_average_precision_compute(state, thresholds_t)


def _multiclass_average_precision_arg_validation(
    num_classes: int,
    average: Optional[str] = "macro",
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
) -> None:
    _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index)
    allowed_average = ("macro", "weighted", "none", None)
    if average not in allowed_average:
        raise ValueError(f"Expected argument `average` to be one of {allowed_average} but got {average}")


def _multiclass_average_precision_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_classes: int,
    average: Optional[str] = "macro",
    thresholds: Optional[Tensor] = None,
) -> Tensor:
    if isinstance(state, Tensor) and thresholds is not None and state.is_cuda and state.ndim == 4:
        from metrics_amd.ops import _hip

        if _hip.hip_available():
            res, weights = _hip.curve_auc_from_confmat(state, mode=1)
            if average in (None, "none"):
                return res
            if average == "macro":
                return res.mean()
            if average == "weighted":
                w = _safe_divide(weights, weights.sum())
                return (res * w).sum()
    precision, recall, _ = _multiclass_precision_recall_curve_compute(state, num_classes, thresholds, average=None)
    return _reduce_average_precision(
        precision,
        recall,
        average,
        weights=_bincount(state[1], minlength=num_classes).float()
        if thresholds is None
        else state[0, :, 1, :].sum(-1),
    )


def multiclass_average_precision(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    average: Optional[str] = "macro",
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Average precision for multiclass tasks."""
    if validate_args:
        _multiclass_average_precision_arg_validation(num_classes, average, thresholds, ignore_index)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None)
    preds_f, target_f, thresholds_t = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        preds_f, target_f, num_classes, thresholds_t, None, ignore_index if not remove_ignored else None
    )
    return _multiclass_average_precision_compute(state, num_classes, average, thresholds_t)


def _multilabel_average_precision_arg_validation(
    num_labels: int,
    average: Optional[str],
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
) -> None:
    _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
    allowed_average = ("micro", "macro", "weighted", "none", None)
    if average not in allowed_average:
        raise ValueError(f"Expected argument `average` to be one of {allowed_average} but got {average}")


def _multilabel_average_precision_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_labels: int,
    average: Optional[str],
    thresholds: Optional[Tensor],
    ignore_index: Optional[int] = None,
) -> Tensor:
    if average == "micro":
        if isinstance(state, Tensor) and thresholds is not None:
            return _binary_average_precision_compute(state.sum(1), thresholds)
        preds = state[0].flatten()
        target = state[1].flatten()
        if ignore_index is not None:
            idx = target != ignore_index
            preds = preds[idx]
            target = target[idx]
        return _binary_average_precision_compute((preds, target), thresholds)

    if isinstance(state, Tensor) and thresholds is not None and state.is_cuda and state.ndim == 4:
        from metrics_amd.ops import _hip

        if _hip.hip_available():
            res, weights = _hip.curve_auc_from_confmat(state, mode=1)
            if average in (None, "none"):
                return res
            if average == "macro":
                return res.mean()
            if average == "weighted":
                w = _safe_divide(weights, weights.sum())
                return (res * w).sum()
    precision, recall, _ = _multilabel_precision_recall_curve_compute(state, num_labels, thresholds, ignore_index)
    return _reduce_average_precision(
        precision,
        recall,
        average,
        weights=(state[1] == 1).sum(dim=0).float() if thresholds is None else state[0, :, 1, :].sum(-1),
    )


def multilabel_average_precision(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    average: Optional[str] = "macro",
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Average precision for multilabel tasks."""
    if validate_args:
        _multilabel_average_precision_arg_validation(num_labels, average, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    preds_f, target_f, thresholds_t = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(
        preds_f, target_f, num_labels, thresholds_t, ignore_index if not remove_ignored else None
    )
    return _multilabel_average_precision_compute(state, num_labels, average, thresholds_t, ignore_index)


def average_precision(
    preds: Tensor,
    target: Tensor,
    task: str,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    average: Optional[str] = "macro",
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Optional[Tensor]:
    """Task-dispatching average precision."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_average_precision(preds, target, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_average_precision(
            preds, target, num_classes, average, thresholds, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_average_precision(
            preds, target, num_labels, average, thresholds, ignore_index, validate_args
        )
    raise ValueError(f"Not handled value: {task}")
