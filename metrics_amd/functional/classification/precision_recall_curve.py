"""Precision-recall curves (binary/multiclass/multilabel).

Parity: torchmetrics ``functional/classification/precision_recall_curve.py``.

Two operating modes (reference semantics preserved):
- ``thresholds=None``: exact curve — preds/target are accumulated (cat state)
  and the curve is a sort + cumsum at compute time.
- ``thresholds`` given (int / list / tensor): constant-memory (T,2,2) confmat
  state. The MI355X update is ONE bucketized-histogram HIP kernel + an
  on-device suffix-sum (csrc/kernels.hip) instead of the reference's
  (N,T) broadcast + bincount (which materializes N*T elements) or its
  per-threshold Python loop.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.utilities.compute import _safe_divide, interp, normalize_logits_if_needed
from metrics_amd.utilities.data import _bincount, _cumsum
from metrics_amd.utilities.enums import ClassificationTask


def _binary_clf_curve(
    preds: Tensor,
    target: Tensor,
    sample_weights: Optional[Union[Sequence, Tensor]] = None,
    pos_label: int = 1,
) -> Tuple[Tensor, Tensor, Tensor]:
    """Cumulative fp/tp counts at each distinct score threshold (descending)."""
    with torch.no_grad():
        if sample_weights is not None and not isinstance(sample_weights, Tensor):
            sample_weights = torch.tensor(sample_weights, device=preds.device, dtype=torch.float)

        # squeeze away a singleton class axis
        if preds.ndim > target.ndim:
            preds = preds[:, 0]

        if preds.is_cuda and preds.numel() > 0:
            # K2 HIP path: rocPRIM radix sort + fused fp64 scans (csrc/clf_curve.hip)
            fps, tps, thr = ops.hip_binary_clf_curve(
                preds, target, sample_weights if isinstance(sample_weights, Tensor) else None, pos_label
            )
            return fps, tps, thr.to(preds.dtype)

        desc_score_indices = torch.argsort(preds, descending=True)

        preds = preds[desc_score_indices]
        target = target[desc_score_indices]

        weight = sample_weights[desc_score_indices] if sample_weights is not None else 1.0

        # pred typically has many tied values. Here we extract the indices
        # associated with the distinct values.
        distinct_value_indices = torch.where(preds[1:] - preds[:-1])[0]
        threshold_idxs = torch.nn.functional.pad(distinct_value_indices, [0, 1], value=target.size(0) - 1)
        target = (target == pos_label).to(torch.long)
        tps = _cumsum(target * weight, dim=0)[threshold_idxs]

        if sample_weights is not None:
            # cumsum keeps fps monotone even under fp rounding of the weights
            fps = _cumsum((1 - target) * weight, dim=0)[threshold_idxs]
        else:
            fps = 1 + threshold_idxs - tps

    return fps, tps, preds[threshold_idxs]


def _adjust_threshold_arg(
    thresholds: Optional[Union[int, List[float], Tensor]] = None, device: Optional[torch.device] = None
) -> Optional[Tensor]:
    """Convert the threshold argument into a tensor (or keep None)."""
    if isinstance(thresholds, int):
        thresholds = torch.linspace(0, 1, thresholds, device=device)
    if isinstance(thresholds, list):
        thresholds = torch.tensor(thresholds, device=device)
    return thresholds


# --------------------------------------------------------------------- binary
def _binary_precision_recall_curve_arg_validation(
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
) -> None:
    if thresholds is not None and not isinstance(thresholds, (list, int, Tensor)):
        raise ValueError(
            "Expected argument `thresholds` to either be an integer, list of floats or"
            f" tensor of floats, but got {thresholds}"
        )
    if isinstance(thresholds, int) and thresholds < 2:
        raise ValueError(f"If argument `thresholds` is an integer, expected it to be larger than 1, but got {thresholds}")
    if isinstance(thresholds, list) and not all(isinstance(t, float) and 0 <= t <= 1 for t in thresholds):
        raise ValueError(f"If argument `thresholds` is a list, expected all elements to be floats in the [0,1] range, but got {thresholds}")
    if isinstance(thresholds, Tensor) and not thresholds.ndim == 1:
        raise ValueError("If argument `thresholds` is an tensor, expected the tensor to be 1d")
    if ignore_index is not None and not isinstance(ignore_index, int):
        raise ValueError(f"Expected argument `ignore_index` to either be `None` or an integer, but got {ignore_index}")


def _binary_precision_recall_curve_tensor_validation(
    preds: Tensor, target: Tensor, ignore_index: Optional[int] = None
) -> None:
    from metrics_amd.utilities.checks import _check_same_shape

    _check_same_shape(preds, target)
    if target.is_floating_point():
        raise ValueError(
            "Expected argument `target` to be an int or long tensor with ground truth labels"
            f" but got tensor with dtype {target.dtype}"
        )
    if not preds.is_floating_point():
        raise ValueError(
            "Expected argument `preds` to be an float tensor with probability/logit scores,"
            f" but got tensor with dtype {preds.dtype}"
        )
    unique_values = torch.unique(target)
    if ignore_index is None:
        check = torch.any((unique_values != 0) & (unique_values != 1))
    else:
        check = torch.any((unique_values != 0) & (unique_values != 1) & (unique_values != ignore_index))
    if check:
        raise RuntimeError(
            f"Detected the following values in `target`: {unique_values} but expected only"
            f" the following values {[0, 1] if ignore_index is None else [ignore_index]}."
        )


def _binary_precision_recall_curve_format(
    preds: Tensor,
    target: Tensor,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    remove_ignored: bool = True,
    normalize: bool = True,
) -> Tuple[Tensor, Tensor, Optional[Tensor]]:
    preds = preds.flatten()
    target = target.flatten()
    if ignore_index is not None and remove_ignored:
        idx = target != ignore_index
        preds = preds[idx]
        target = target[idx]

    if normalize:
        preds = normalize_logits_if_needed(preds, "sigmoid")
    thresholds = _adjust_threshold_arg(thresholds, preds.device)
    return preds, target, thresholds


def _binary_precision_recall_curve_update(
    preds: Tensor,
    target: Tensor,
    thresholds: Optional[Tensor],
    ignore_index: Optional[int] = None,
) -> Union[Tensor, Tuple[Tensor, Tensor]]:
    """Return (preds, target) when thresholds is None, else the (T,2,2) confmat delta."""
    if thresholds is None:
        return preds, target
    return ops.binary_curve_confmat(preds, target, thresholds, ignore_index)


def _binary_precision_recall_curve_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    thresholds: Optional[Tensor],
    pos_label: int = 1,
    precomputed: Optional[Tuple[Tensor, Tensor, Tensor]] = None,
) -> Tuple[Tensor, Tensor, Tensor]:
    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, 1, 1]
        fps = state[:, 0, 1]
        fns = state[:, 1, 0]
        precision = _safe_divide(tps, tps + fps)
        recall = _safe_divide(tps, tps + fns)
        precision = torch.cat([precision, torch.ones(1, dtype=precision.dtype, device=precision.device)])
        recall = torch.cat([recall, torch.zeros(1, dtype=recall.dtype, device=recall.device)])
        return precision, recall, thresholds

    if precomputed is not None:
        fps, tps, thresholds = precomputed[:3]
    else:
        fps, tps, thresholds = _binary_clf_curve(state[0], state[1], pos_label=pos_label)
    precision = tps / (tps + fps)
    recall = tps / tps[-1]

    # flip() rather than a negative-stride slice (torch has no negative strides)
    precision = torch.cat([reversed(precision), torch.ones(1, dtype=precision.dtype, device=precision.device)])
    recall = torch.cat([reversed(recall), torch.zeros(1, dtype=recall.dtype, device=recall.device)])
    thresholds = reversed(thresholds).detach().clone()
    return precision, recall, thresholds


def binary_precision_recall_curve(
    preds: Tensor,
    target: Tensor,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tuple[Tensor, Tensor, Tensor]:
    """PR curve for binary tasks; returns (precision, recall, thresholds)."""
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
    return _binary_precision_recall_curve_compute(state, thresholds_t)


# ------------------------------------------------------------------ multiclass
def _multiclass_precision_recall_curve_arg_validation(
    num_classes: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    average: Optional[str] = None,
) -> None:
    if not isinstance(num_classes, int) or num_classes < 2:
        raise ValueError(f"Expected argument `num_classes` to be an integer larger than 1, but got {num_classes}")
    if average not in (None, "micro", "macro"):
        raise ValueError(f"Expected argument `average` to be one of None, 'micro' or 'macro', but got {average}")
    _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)


def _multiclass_precision_recall_curve_tensor_validation(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int] = None
) -> None:
    if not preds.ndim == target.ndim + 1:
        raise ValueError(
            f"Expected `preds` to have one more dimension than `target` but got {preds.ndim} and {target.ndim}"
        )
    if target.is_floating_point():
        raise ValueError(f"Expected argument `target` to be an int or long tensor, but got {target.dtype}")
    if not preds.is_floating_point():
        raise ValueError(f"Expected `preds` to contain floating point values, but got values with dtype {preds.dtype}")
    if preds.shape[1] != num_classes:
        raise ValueError(
            f"Expected `preds.shape[1]={preds.shape[1]}` to be equal to the number of classes {num_classes}"
        )
    if preds.shape[2:] != target.shape[1:]:
        raise ValueError("Expected the shape of `preds` should be (N, C, ...) and the shape of `target` should be (N, ...).")

    num_unique_values = len(torch.unique(target))
    check = num_unique_values > num_classes if ignore_index is None else num_unique_values > num_classes + 1
    if check:
        raise RuntimeError(
            "Detected more unique values in `target` than expected. Expected only"
            f" {num_classes if ignore_index is None else num_classes + 1} but found {num_unique_values}"
        )


def _multiclass_precision_recall_curve_format(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    average: Optional[str] = None,
    remove_ignored: bool = True,
    normalize: bool = True,
) -> Tuple[Tensor, Tensor, Optional[Tensor]]:
    if preds.ndim == 2:
        pass  # already (N, C); avoid the transpose->reshape round-trip copy
    else:
        preds = preds.transpose(0, 1).reshape(num_classes, -1).T
    target = target.flatten()

    if ignore_index is not None and remove_ignored:
        idx = target != ignore_index
        preds = preds[idx]
        target = target[idx]

    if normalize:
        preds = normalize_logits_if_needed(preds, "softmax")

    if average == "micro":
        preds = preds.flatten()
        target = torch.nn.functional.one_hot(target.clamp(min=0), num_classes=num_classes).flatten()

    thresholds = _adjust_threshold_arg(thresholds, preds.device)
    return preds, target, thresholds


def _multiclass_precision_recall_curve_update(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    thresholds: Optional[Tensor],
    average: Optional[str] = None,
    ignore_index: Optional[int] = None,
) -> Union[Tensor, Tuple[Tensor, Tensor]]:
    if thresholds is None:
        return preds, target
    if average == "micro":
        return _binary_precision_recall_curve_update(preds, target, thresholds, ignore_index)
    return ops.multiclass_curve_confmat(preds, target, thresholds, ignore_index)


def _multiclass_precision_recall_curve_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_classes: int,
    thresholds: Optional[Tensor],
    average: Optional[str] = None,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    if average == "micro":
        return _binary_precision_recall_curve_compute(state, thresholds)

    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, :, 1, 1]
        fps = state[:, :, 0, 1]
        fns = state[:, :, 1, 0]
        precision = _safe_divide(tps, tps + fps)
        recall = _safe_divide(tps, tps + fns)
        precision = torch.cat([precision, torch.ones(1, num_classes, dtype=precision.dtype, device=precision.device)])
        recall = torch.cat([recall, torch.zeros(1, num_classes, dtype=recall.dtype, device=recall.device)])
        precision = precision.T
        recall = recall.T
        thres = thresholds
        tensor_state = True
    else:
        precision_list, recall_list, thres_list = [], [], []
        # GPU: ONE composite-key sort yields every class's curve (K2 batched)
        curves = (
            ops.hip_mc_clf_curve(state[0], state[1])
            if state[0].is_cuda and state[0].numel()
            else [None] * num_classes
        )
        for i in range(num_classes):
            res = _binary_precision_recall_curve_compute(
                (state[0][:, i], state[1]), thresholds=None, pos_label=i, precomputed=curves[i]
            )
            precision_list.append(res[0])
            recall_list.append(res[1])
            thres_list.append(res[2])
        tensor_state = False

    if average == "macro":
        thres = thres.repeat(num_classes) if tensor_state else torch.cat(thres_list, 0)
        thres = thres.sort().values
        mean_precision = precision.flatten() if tensor_state else torch.cat(precision_list, 0)
        mean_precision = mean_precision.sort().values
        mean_recall = torch.zeros_like(mean_precision)
        for i in range(num_classes):
            p = precision[i] if tensor_state else precision_list[i]
            r = recall[i] if tensor_state else recall_list[i]
            mean_recall += interp(mean_precision, p.flip(0), r.flip(0))
        mean_recall /= num_classes
        return mean_precision, mean_recall, thres

    if tensor_state:
        return precision, recall, thres
    return precision_list, recall_list, thres_list


def multiclass_precision_recall_curve(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    average: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    """PR curves for multiclass tasks (one-vs-rest)."""
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
    return _multiclass_precision_recall_curve_compute(state, num_classes, thresholds_t, average)


# ------------------------------------------------------------------ multilabel
def _multilabel_precision_recall_curve_arg_validation(
    num_labels: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
) -> None:
    if not isinstance(num_labels, int) or num_labels < 2:
        raise ValueError(f"Expected argument `num_labels` to be an integer larger than 1, but got {num_labels}")
    _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)


def _multilabel_precision_recall_curve_tensor_validation(
    preds: Tensor, target: Tensor, num_labels: int, ignore_index: Optional[int] = None
) -> None:
    from metrics_amd.utilities.checks import _check_same_shape

    _check_same_shape(preds, target)
    if target.is_floating_point():
        raise ValueError(f"Expected argument `target` to be an int or long tensor, but got {target.dtype}")
    if not preds.is_floating_point():
        raise ValueError(f"Expected `preds` to contain floating point values, but got values with dtype {preds.dtype}")
    if preds.shape[1] != num_labels:
        raise ValueError(
            f"Expected `preds.shape[1]={preds.shape[1]}` to be equal to the number of labels {num_labels}"
        )


def _multilabel_precision_recall_curve_format(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    remove_ignored: bool = True,
    normalize: bool = True,
) -> Tuple[Tensor, Tensor, Optional[Tensor]]:
    if preds.ndim == 2:
        pass  # already (N, L)
    else:
        preds = preds.transpose(0, 1).reshape(num_labels, -1).T
        target = target.transpose(0, 1).reshape(num_labels, -1).T

    if normalize:
        preds = normalize_logits_if_needed(preds, "sigmoid")

    thresholds = _adjust_threshold_arg(thresholds, preds.device)
    if ignore_index is not None and thresholds is not None and remove_ignored:
        preds = preds.clone()
        target = target.clone()
        # sentinel must stay negative after every additive offset in the
        # bincount mapping: -4*L*T dominates the max offset 4*L*(T-1)+4(L-1)+1
        idx = target == ignore_index
        sentinel = -4 * num_labels * (len(thresholds) if thresholds is not None else 1)
        preds[idx] = float(sentinel)
        target[idx] = sentinel
    return preds, target, thresholds


def _multilabel_precision_recall_curve_update(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    thresholds: Optional[Tensor],
    ignore_index: Optional[int] = None,
) -> Union[Tensor, Tuple[Tensor, Tensor]]:
    if thresholds is None:
        return preds, target
    return ops.multilabel_curve_confmat(preds, target, thresholds, ignore_index)


def _multilabel_precision_recall_curve_compute(
    state: Union[Tensor, Tuple[Tensor, Tensor]],
    num_labels: int,
    thresholds: Optional[Tensor],
    ignore_index: Optional[int] = None,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    if isinstance(state, Tensor) and thresholds is not None:
        tps = state[:, :, 1, 1]
        fps = state[:, :, 0, 1]
        fns = state[:, :, 1, 0]
        precision = _safe_divide(tps, tps + fps)
        recall = _safe_divide(tps, tps + fns)
        precision = torch.cat([precision, torch.ones(1, num_labels, dtype=precision.dtype, device=precision.device)])
        recall = torch.cat([recall, torch.zeros(1, num_labels, dtype=recall.dtype, device=recall.device)])
        return precision.T, recall.T, thresholds

    precision_list, recall_list, thres_list = [], [], []
    curves = (
        ops.hip_mc_clf_curve(state[0], state[1], multilabel=True)
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
        res = _binary_precision_recall_curve_compute(
            (preds, target), thresholds=None, pos_label=1, precomputed=curves[i]
        )
        precision_list.append(res[0])
        recall_list.append(res[1])
        thres_list.append(res[2])
    return precision_list, recall_list, thres_list


def multilabel_precision_recall_curve(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    thresholds: Optional[Union[int, List[float], Tensor]] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
    """PR curves for multilabel tasks (per label)."""
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
    return _multilabel_precision_recall_curve_compute(state, num_labels, thresholds_t, ignore_index)


def precision_recall_curve(
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
    """Task-dispatching precision-recall curve."""
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return binary_precision_recall_curve(preds, target, thresholds, ignore_index, validate_args)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_precision_recall_curve(
            preds, target, num_classes, thresholds, average, ignore_index, validate_args
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return multilabel_precision_recall_curve(preds, target, num_labels, thresholds, ignore_index, validate_args)
    raise ValueError(f"Not handled value: {task}")
