"""Modular PR curves. Parity: torchmetrics ``classification/precision_recall_curve.py``.

State layout matches the reference exactly: cat-list states ``preds``/``target``
when ``thresholds is None``; a constant-memory ``confmat`` (T,2,2)/(T,C,2,2)/
(T,L,2,2) long state otherwise (GPU update: bucketized-histogram HIP kernel).
"""
from __future__ import annotations

from typing import Any, List, Optional, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.precision_recall_curve import (
    _adjust_threshold_arg,
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


class BinaryPrecisionRecallCurve(Metric):
    """PR curve for binary tasks (stateful)."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False
    preds: List[Tensor]
    target: List[Tensor]
    confmat: Tensor

    def __init__(
        self,
        thresholds: Optional[Union[int, List[float], Tensor]] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _binary_precision_recall_curve_arg_validation(thresholds, ignore_index)
        self.ignore_index = ignore_index
        self.validate_args = validate_args

        thresholds = _adjust_threshold_arg(thresholds)
        if thresholds is None:
            self.thresholds = thresholds
            self.add_state("preds", default=[], dist_reduce_fx="cat")
            self.add_state("target", default=[], dist_reduce_fx="cat")
        else:
            self.register_buffer("thresholds", thresholds, persistent=False)
            self.add_state(
                "confmat", default=torch.zeros(len(thresholds), 2, 2, dtype=torch.long), dist_reduce_fx="sum"
            )

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate either (preds, target) lists or the (T,2,2) confmat."""
        if self.validate_args:
            _binary_precision_recall_curve_tensor_validation(preds, target, self.ignore_index)
        gpu_fast = preds.is_cuda and self.thresholds is not None and preds.dtype in (torch.float32, torch.bfloat16)
        remove_ignored = not gpu_fast
        preds_f, target_f, _ = _binary_precision_recall_curve_format(
            preds, target, self.thresholds, self.ignore_index, remove_ignored=remove_ignored,
            normalize=not gpu_fast,  # kernel applies sigmoid-iff-logits in-flight
        )
        if gpu_fast:
            from metrics_amd.ops import _hip

            _hip.curve_hist_into_confmat(
                preds_f, target_f, self.thresholds, self.ignore_index, self.confmat, mode=0, norm="sigmoid", owner=self
            )
            return
        state = _binary_precision_recall_curve_update(
            preds_f, target_f, self.thresholds, self.ignore_index if not remove_ignored else None
        )
        if isinstance(state, Tensor):
            self.confmat += state
        else:
            self.preds.append(state[0])
            self.target.append(state[1])

    def compute(self) -> Tuple[Tensor, Tensor, Tensor]:
        """(precision, recall, thresholds)."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _binary_precision_recall_curve_compute(state, self.thresholds)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        # plot convention: recall on x, precision on y
        curve_computed = (curve_computed[1], curve_computed[0], curve_computed[2])
        score = (
            _auc_score(curve_computed[0], curve_computed[1]) if score is True else None if score is False else score
        )
        return plot_curve(
            curve_computed, score=score, ax=ax, label_names=("Recall", "Precision"), name=self.__class__.__name__
        )


def _auc_score(x: Tensor, y: Tensor) -> Tensor:
    from metrics_amd.utilities.compute import _auc_compute_without_check

    return _auc_compute_without_check(x, y, 1.0)


class MulticlassPrecisionRecallCurve(Metric):
    _hip_fused_kind = "mc_curve"
    """PR curves for multiclass tasks (stateful)."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False
    preds: List[Tensor]
    target: List[Tensor]
    confmat: Tensor

    def __init__(
        self,
        num_classes: int,
        thresholds: Optional[Union[int, List[float], Tensor]] = None,
        average: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _multiclass_precision_recall_curve_arg_validation(num_classes, thresholds, ignore_index, average)
        self.num_classes = num_classes
        self.average = average
        self.ignore_index = ignore_index
        self.validate_args = validate_args

        thresholds = _adjust_threshold_arg(thresholds)
        if thresholds is None:
            self.thresholds = thresholds
            self.add_state("preds", default=[], dist_reduce_fx="cat")
            self.add_state("target", default=[], dist_reduce_fx="cat")
        else:
            self.register_buffer("thresholds", thresholds, persistent=False)
            size = (len(thresholds), num_classes, 2, 2) if average != "micro" else (len(thresholds), 2, 2)
            self.add_state("confmat", default=torch.zeros(*size, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate either (preds, target) lists or the (T,C,2,2) confmat."""
        if self.validate_args:
            _multiclass_precision_recall_curve_tensor_validation(preds, target, self.num_classes, self.ignore_index)
        gpu_fast = (
            preds.is_cuda
            and self.thresholds is not None
            and self.average != "micro"
            and preds.dtype in (torch.float32, torch.bfloat16)
        )
        remove_ignored = not gpu_fast
        preds_f, target_f, _ = _multiclass_precision_recall_curve_format(
            preds, target, self.num_classes, self.thresholds, self.ignore_index, self.average,
            remove_ignored=remove_ignored,
            normalize=not gpu_fast,  # kernel applies softmax-iff-logits in-flight
        )
        if gpu_fast:
            from metrics_amd.ops import _hip

            _hip.curve_hist_into_confmat(
                preds_f, target_f, self.thresholds, self.ignore_index if not remove_ignored else None,
                self.confmat, mode=0, norm="softmax", owner=self,
            )
            return
        state = _multiclass_precision_recall_curve_update(
            preds_f, target_f, self.num_classes, self.thresholds, self.average,
            self.ignore_index if not remove_ignored else None,
        )
        if isinstance(state, Tensor):
            self.confmat += state
        else:
            self.preds.append(state[0])
            self.target.append(state[1])

    def compute(self) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
        """(precision, recall, thresholds) per class."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _multiclass_precision_recall_curve_compute(state, self.num_classes, self.thresholds, self.average)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        curve_computed = (curve_computed[1], curve_computed[0], curve_computed[2])
        return plot_curve(
            curve_computed, score=score if score is not True else None, ax=ax,
            label_names=("Recall", "Precision"), name=self.__class__.__name__,
        )


class MultilabelPrecisionRecallCurve(Metric):
    """PR curves for multilabel tasks (stateful)."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False
    preds: List[Tensor]
    target: List[Tensor]
    confmat: Tensor

    def __init__(
        self,
        num_labels: int,
        thresholds: Optional[Union[int, List[float], Tensor]] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _multilabel_precision_recall_curve_arg_validation(num_labels, thresholds, ignore_index)
        self.num_labels = num_labels
        self.ignore_index = ignore_index
        self.validate_args = validate_args

        thresholds = _adjust_threshold_arg(thresholds)
        if thresholds is None:
            self.thresholds = thresholds
            self.add_state("preds", default=[], dist_reduce_fx="cat")
            self.add_state("target", default=[], dist_reduce_fx="cat")
        else:
            self.register_buffer("thresholds", thresholds, persistent=False)
            self.add_state(
                "confmat", default=torch.zeros(len(thresholds), num_labels, 2, 2, dtype=torch.long),
                dist_reduce_fx="sum",
            )

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate either (preds, target) lists or the (T,L,2,2) confmat."""
        if self.validate_args:
            _multilabel_precision_recall_curve_tensor_validation(preds, target, self.num_labels, self.ignore_index)
        gpu_fast = preds.is_cuda and self.thresholds is not None and preds.dtype in (torch.float32, torch.bfloat16)
        remove_ignored = not preds.is_cuda
        preds_f, target_f, _ = _multilabel_precision_recall_curve_format(
            preds, target, self.num_labels, self.thresholds, self.ignore_index, remove_ignored=remove_ignored,
            normalize=not gpu_fast,  # kernel applies sigmoid-iff-logits in-flight
        )
        if gpu_fast:
            from metrics_amd.ops import _hip

            _hip.curve_hist_into_confmat(
                preds_f, target_f, self.thresholds, self.ignore_index if not remove_ignored else None,
                self.confmat, mode=1, norm="sigmoid", owner=self,
            )
            return
        state = _multilabel_precision_recall_curve_update(
            preds_f, target_f, self.num_labels, self.thresholds, self.ignore_index if not remove_ignored else None
        )
        if isinstance(state, Tensor):
            self.confmat += state
        else:
            self.preds.append(state[0])
            self.target.append(state[1])

    def compute(self) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
        """(precision, recall, thresholds) per label."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _multilabel_precision_recall_curve_compute(state, self.num_labels, self.thresholds, self.ignore_index)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        curve_computed = (curve_computed[1], curve_computed[0], curve_computed[2])
        return plot_curve(
            curve_computed, score=score if score is not True else None, ax=ax,
            label_names=("Recall", "Precision"), name=self.__class__.__name__,
        )


class PrecisionRecallCurve(_ClassificationTaskWrapper):
    """Task-dispatching PrecisionRecallCurve."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        thresholds: Optional[Union[int, List[float], Tensor]] = None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        average: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        kwargs.update({"thresholds": thresholds, "ignore_index": ignore_index, "validate_args": validate_args})
        if task == ClassificationTask.BINARY:
            return BinaryPrecisionRecallCurve(**kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassPrecisionRecallCurve(num_classes, average=average, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelPrecisionRecallCurve(num_labels, **kwargs)
        raise ValueError(f"Not handled value: {task}")
