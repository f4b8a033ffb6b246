"""ROC curves. Parity: torchmetrics ``functional/classification/roc.py``."""
from __future__ import annotations

from typing import List, Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.compute import _safe_divide, interp
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.utilities.prints import rank_zero_warn
from metrics_amd.functional.classification.precision_recall_curve import (
    _binary_clf_curve,
    _binary_precision_recall_curve_arg_validation,
    _binary_precision_recall_curve_format,
    _binary_precision_recall_curve_tensor_validation,
    _binary_precision_recall_curve_update,
    _multiclass_precision_recall_curve_arg_validation,
    _multiclass_precision_recall_curve_format,
    _multiclass_precision_recall_curve_tensor_validation,
    _multiclass_precision_recall_curve_update,
    _multilabel_precision_recall_curve_arg_validation,
    _multilabel_precision_recall_curve_format,
    _multilabel_precision_recall_curve_tensor_validation,
    _multilabel_precision_recall_curve_update,
)


def _binary_roc_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    thresholds: Optional[Tensor],
    pos_label: int = 1,
    precomputed=None,
) -> Tuple[Tensor, Tensor, Tensor]:
    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, 1, 1]
        fps = state[:, 0, 1]
        fns = state[:, 1, 0]
        tns = state[:, 0, 0]
        tpr = _safe_divide(tps, tps + fns).flip(0)
        fpr = _safe_divide(fps, fps + tns).flip(0)
        thres = thresholds.flip(0)
    else:
        known_pos = known_neg = None
        if precomputed is not None:
            fps, tps, thres, known_pos, known_neg = precomputed
        else:
            fps, tps, thres = _binary_clf_curve(preds=state[0], target=state[1], pos_label=pos_label)
        # add extra threshold position so that the curve starts at (0, 0)
        tps = torch.cat([torch.zeros(1, dtype=tps.dtype, device=tps.device), tps])
        fps = torch.cat([torch.zeros(1, dtype=fps.dtype, device=fps.device), fps])
        thres = torch.cat([torch.ones(1, dtype=thres.dtype, device=thres.device), thres])

        # fps[-1] == #negatives and tps[-1] == #positives: with the batched
        # curve path those counts are already on the host — no device sync
        if (known_neg <= 0) if known_neg is not None else (fps[-1] <= 0):
            rank_zero_warn(
                "No negative samples in targets, false positive value should be meaningless."
                " Returning zero tensor in false positive score",
                UserWarning,
            )
            fpr = torch.zeros_like(thres)
        else:
            fpr = fps / fps[-1]

        if (known_pos <= 0) if known_pos is not None else (tps[-1] <= 0):
            rank_zero_warn(
                "No positive samples in targets, true positive value should be meaningless."
                " Returning zero tensor in true positive score",
                UserWarning,
            )
            tpr = torch.zeros_like(thres)
        else:
            tpr = tps / tps[-1]

    return fpr, tpr, thres


def binary_roc(
    preds: Tensor,
    target: Tensor,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor, Tensor]:
    """ROC for binary tasks; returns (fpr, tpr, thresholds)."""
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
    return _binary_roc_compute(state, thresholds_t)


def _multiclass_roc_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_classes: int,
    thresholds: Optional[Tensor],
    average: Optional[str] = None,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    if average == "micro":
        return _binary_roc_compute(state, thresholds, pos_label=1)

    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, :, 1, 1]
        fps = state[:, :, 0, 1]
        fns = state[:, :, 1, 0]
        tns = state[:, :, 0, 0]
        tpr = _safe_divide(tps, tps + fns).flip(0).T
        fpr = _safe_divide(fps, fps + tns).flip(0).T
        thres = thresholds.flip(0)
        tensor_state = True
    else:
        fpr_list, tpr_list, thres_list = [], [], []
        from metrics_amd import ops as _ops

        curves = (
            _ops.hip_mc_clf_curve(state[0], state[1])
            if state[0].is_cuda and state[0].numel()
            else [None] * num_classes
        )
        for i in range(num_classes):
            res = _binary_roc_compute(
                (state[0][:, i], state[1]), thresholds=None, pos_label=i, precomputed=curves[i]
            )
            fpr_list.append(res[0])
            tpr_list.append(res[1])
            thres_list.append(res[2])
        tensor_state = False

    if average == "macro":
        thres = thres.repeat(num_classes) if tensor_state else torch.cat(thres_list, 0)
        thres = thres.sort(descending=True).values
        mean_fpr = fpr.flatten() if tensor_state else torch.cat(fpr_list, 0)
        mean_fpr = mean_fpr.sort().values
        mean_tpr = torch.zeros_like(mean_fpr)
        for i in range(num_classes):
            f = fpr[i] if tensor_state else fpr_list[i]
            t = tpr[i] if tensor_state else tpr_list[i]
            mean_tpr += interp(mean_fpr, f, t)
        mean_tpr /= num_classes
        return mean_fpr, mean_tpr, thres

    if tensor_state:
        return fpr, tpr, thres
    return fpr_list, tpr_list, thres_list


def multiclass_roc(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    average: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    """ROC for multiclass tasks (one-vs-rest)."""
    if validate_args:
        _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index, average)
        _multiclass_precision_recall_curve_tensor_validation(preds, target, num_classes, ignore_index)
    remove_ignored = not (preds.is_cuda and thresholds is not None and average != "micro")
    preds_f, target_f, thresholds_t = _multiclass_precision_recall_curve_format(
        preds, target, num_classes, thresholds, ignore_index, average, remove_ignored=remove_ignored
    )
    state = _multiclass_precision_recall_curve_update(
        preds_f, target_f, num_classes, thresholds_t, average, ignore_index if not remove_ignored else None
    )
    return _multiclass_roc_compute(state, num_classes, thresholds_t, average)


def _multilabel_roc_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_labels: int,
    thresholds: Optional[Tensor],
    ignore_index: Optional[int] = None,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, :, 1, 1]
        fps = state[:, :, 0, 1]
        fns = state[:, :, 1, 0]
        tns = state[:, :, 0, 0]
        tpr = _safe_divide(tps, tps + fns).flip(0).T
        fpr = _safe_divide(fps, fps + tns).flip(0).T
        thres = thresholds.flip(0)
        return fpr, tpr, thres

    fpr_list, tpr_list, thres_list = [], [], []
    from metrics_amd import ops as _ops

    curves = (
        _ops.hip_mc_clf_curve(state[0], state[1], multilabel=True)
        if state[0].is_cuda and state[0].numel() and ignore_index is None
        else [None] * num_labels
    )
    for i in range(num_labels):
        preds = state[0][:, i]
        target = state[1][:, i]
        if ignore_index is not None:
            idx = target != ignore_index
            preds = preds[idx]
            target = target[idx]
        res = _binary_roc_compute((preds, target), thresholds=None, pos_label=1, precomputed=curves[i])
        fpr_list.append(res[0])
        tpr_list.append(res[1])
        thres_list.append(res[2])
    return fpr_list, tpr_list, thres_list


def multilabel_roc(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    """ROC for multilabel tasks (per label)."""
    if validate_args:
        _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        _multilabel_precision_recall_curve_tensor_validation(preds, target, num_labels, ignore_index)
    remove_ignored = not preds.is_cuda
    preds_f, target_f, thresholds_t = _multilabel_precision_recall_curve_format(
        preds, target, num_labels, thresholds, ignore_index, remove_ignored=remove_ignored
    )
    state = _multilabel_precision_recall_curve_update(
        preds_f, target_f, num_labels, thresholds_t, ignore_index if not remove_ignored else None
    )
    return _multilabel_roc_compute(state, num_labels, thresholds_t, ignore_index)


def roc(
    preds: Tensor,
    target: Tensor,
    task: str,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    average: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    """Task-dispatching ROC."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_roc(preds, target, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_roc(preds, target, num_classes, thresholds, average, ignore_index, validate_args)
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_roc(preds, target, num_labels, thresholds, ignore_index, validate_args)
    raise ValueError(f"Not handled value: {task}")
