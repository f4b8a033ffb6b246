"""Precision & Recall. Parity: torchmetrics ``functional/classification/precision_recall.py``."""
from __future__ import annotations

from typing import Optional

from torch import Tensor

from metrics_amd.utilities.compute import _adjust_weights_safe_divide, _safe_divide
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.stat_scores import (
    _binary_stat_scores_arg_validation,
    _binary_stat_scores_pipeline,
    _binary_stat_scores_tensor_validation,
    _multiclass_stat_scores_arg_validation,
    _multiclass_stat_scores_pipeline,
    _multiclass_stat_scores_tensor_validation,
    _multilabel_stat_scores_arg_validation,
    _multilabel_stat_scores_pipeline,
    _multilabel_stat_scores_tensor_validation,
)


def _precision_recall_reduce(
    stat: str,
    tp: Tensor,
    fp: Tensor,
    tn: Tensor,
    fn: Tensor,
    average: Optional[str],
    multidim_average: str = "global",
    multilabel: bool = False,
    top_k: int = 1,
    zero_division: float = 0,
) -> Tensor:
    different_stat = fp if stat == "precision" else fn  # this is what differs between the two scores
    if (
        tp.is_cuda and tp.ndim == 1 and multidim_average == "global" and not multilabel
        and average in ("micro", "macro", "weighted")
    ):
        from metrics_amd.ops import _hip

        if _hip.hip_available():
            den = (1, 1, 0, 0) if stat == "precision" else (1, 0, 0, 1)
            return _hip.linear_stat_compute(
                tp, fp, tn, fn, (1, 0, 0, 0), den, average, top_k != 1, zero_division
            )
    if average == "binary":
        return _safe_divide(tp, tp + different_stat, zero_division)
    if average == "micro":
        tp = tp.sum(dim=0 if multidim_average == "global" else 1)
        different_stat = different_stat.sum(dim=0 if multidim_average == "global" else 1)
        return _safe_divide(tp, tp + different_stat, zero_division)

    score = _safe_divide(tp, tp + different_stat, zero_division)
    return _adjust_weights_safe_divide(score, average, multilabel, tp, fp, fn, top_k)


def _make_task_fns(stat: str):
    def binary_fn(
        preds: Tensor,
        target: Tensor,
        threshold: float = 0.5,
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0,
    ) -> Tensor:
        if validate_args:
            _binary_stat_scores_arg_validation(threshold, multidim_average, ignore_index, zero_division)
            _binary_stat_scores_tensor_validation(preds, target, multidim_average, ignore_index)
        tp, fp, tn, fn = _binary_stat_scores_pipeline(preds, target, threshold, multidim_average, ignore_index)
        return _precision_recall_reduce(
            stat, tp, fp, tn, fn, average="binary", multidim_average=multidim_average, zero_division=zero_division
        )

    def multiclass_fn(
        preds: Tensor,
        target: Tensor,
        num_classes: int,
        average: Optional[str] = "macro",
        top_k: int = 1,
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0,
    ) -> Tensor:
        if validate_args:
            _multiclass_stat_scores_arg_validation(
                num_classes, top_k, average, multidim_average, ignore_index, zero_division
            )
            _multiclass_stat_scores_tensor_validation(preds, target, num_classes, multidim_average, ignore_index)
        tp, fp, tn, fn = _multiclass_stat_scores_pipeline(
            preds, target, num_classes, top_k, average, multidim_average, ignore_index
        )
        return _precision_recall_reduce(
            stat, tp, fp, tn, fn, average=average, multidim_average=multidim_average, top_k=top_k,
            zero_division=zero_division,
        )

    def multilabel_fn(
        preds: Tensor,
        target: Tensor,
        num_labels: int,
        threshold: float = 0.5,
        average: Optional[str] = "macro",
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0,
    ) -> Tensor:
        if validate_args:
            _multilabel_stat_scores_arg_validation(
                num_labels, threshold, average, multidim_average, ignore_index, zero_division
            )
            _multilabel_stat_scores_tensor_validation(preds, target, num_labels, multidim_average, ignore_index)
        tp, fp, tn, fn = _multilabel_stat_scores_pipeline(
            preds, target, num_labels, threshold, multidim_average, ignore_index
        )
        return _precision_recall_reduce(
            stat, tp, fp, tn, fn, average=average, multidim_average=multidim_average, multilabel=True,
            zero_division=zero_division,
        )

    return binary_fn, multiclass_fn, multilabel_fn


binary_precision, multiclass_precision, multilabel_precision = _make_task_fns("precision")
binary_recall, multiclass_recall, multilabel_recall = _make_task_fns("recall")
binary_precision.__name__ = "binary_precision"
multiclass_precision.__name__ = "multiclass_precision"
multilabel_precision.__name__ = "multilabel_precision"
binary_recall.__name__ = "binary_recall"
multiclass_recall.__name__ = "multiclass_recall"
multilabel_recall.__name__ = "multilabel_recall"


def _dispatch(
    stat: str,
    preds: Tensor,
    target: Tensor,
    task: str,
    threshold: float,
    num_classes: Optional[int],
    num_labels: Optional[int],
    average: Optional[str],
    multidim_average: str,
    top_k: int,
    ignore_index: Optional[int],
    validate_args: bool,
    zero_division: float = 0,
) -> Tensor:
    b, mc, ml = (binary_precision, multiclass_precision, multilabel_precision) if stat == "precision" else (
        binary_recall, multiclass_recall, multilabel_recall
    )
    task = ClassificationTask.from_str(task)
    if task == ClassificationTask.BINARY:
        return b(preds, target, threshold, multidim_average, ignore_index, validate_args, zero_division)
    if task == ClassificationTask.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        if not isinstance(top_k, int):
            raise ValueError(f"`top_k` is expected to be `int` but `{type(top_k)} was passed.`")
        return mc(
            preds, target, num_classes, average, top_k, multidim_average, ignore_index, validate_args, zero_division
        )
    if task == ClassificationTask.MULTILABEL:
        if not isinstance(num_labels, int):
            raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
        return ml(
            preds, target, num_labels, threshold, average, multidim_average, ignore_index, validate_args, zero_division
        )
    raise ValueError(f"Not handled value: {task}")


def precision(
    preds: Tensor,
    target: Tensor,
    task: str,
    threshold: float = 0.5,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    average: Optional[str] = "micro",
    multidim_average: str = "global",
    top_k: int = 1,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
    zero_division: float = 0,
) -> Tensor:
    """Task-dispatching precision."""
    return _dispatch(
        "precision", preds, target, task, threshold, num_classes, num_labels, average, multidim_average, top_k,
        ignore_index, validate_args, zero_division,
    )


def recall(
    preds: Tensor,
    target: Tensor,
    task: str,
    threshold: float = 0.5,
    num_classes: Optional[int] = None,
    num_labels: Optional[int] = None,
    average: Optional[str] = "micro",
    multidim_average: str = "global",
    top_k: int = 1,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
    zero_division: float = 0,
) -> Tensor:
    """Task-dispatching recall."""
    return _dispatch(
        "recall", preds, target, task, threshold, num_classes, num_labels, average, multidim_average, top_k,
        ignore_index, validate_args, zero_division,
    )
