"""Modular Jaccard index. Parity: torchmetrics ``classification/jaccard.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.jaccard import _jaccard_index_reduce
from metrics_amd.functional.classification.confusion_matrix import (
    _binary_confusion_matrix_arg_validation,
    _binary_confusion_matrix_format,
    _binary_confusion_matrix_tensor_validation,
    _binary_confusion_matrix_update,
    _multiclass_confusion_matrix_arg_validation,
    _multiclass_confusion_matrix_format,
    _multiclass_confusion_matrix_tensor_validation,
    _multiclass_confusion_matrix_update,
    _multilabel_confusion_matrix_arg_validation,
    _multilabel_confusion_matrix_format,
    _multilabel_confusion_matrix_tensor_validation,
    _multilabel_confusion_matrix_update,
)


class BinaryJaccardIndex(Metric):
    """Jaccard index for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0
    confmat: Tensor

    def __init__(
        self,
        threshold: float = 0.5,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0.0,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.zero_division = zero_division
        if validate_args:
            _binary_confusion_matrix_arg_validation(threshold, ignore_index, normalize=None)
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.validate_args = validate_args

        self.add_state("confmat", torch.zeros(2, 2, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrix."""
        if self.validate_args:
            _binary_confusion_matrix_tensor_validation(preds, target, self.ignore_index)
        preds, target = _binary_confusion_matrix_format(preds, target, self.threshold, self.ignore_index)
        self.confmat += _binary_confusion_matrix_update(preds, target)

    def compute(self) -> Tensor:
        return _jaccard_index_reduce(self.confmat, average="binary", zero_division=self.zero_division)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MulticlassJaccardIndex(Metric):
    _hip_fused_kind = "mc_confmat"
    """Jaccard index for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0
    plot_legend_name: str = "Class"
    confmat: Tensor

    def __init__(
        self,
        num_classes: int,
        average: Optional[str] = "macro",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0.0,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.zero_division = zero_division
        if validate_args:
            _multiclass_confusion_matrix_arg_validation(num_classes, ignore_index, normalize=None)
        self.num_classes = num_classes
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.average = average

        self.add_state("confmat", torch.zeros(num_classes, num_classes, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrix (GPU: fused HIP kernel)."""
        if self.validate_args:
            _multiclass_confusion_matrix_tensor_validation(preds, target, self.num_classes, self.ignore_index)
        from metrics_amd.classification.confusion_matrix import _fused_confmat_update

        if _fused_confmat_update(self, preds, target):
            return
        preds, target = _multiclass_confusion_matrix_format(preds, target, self.ignore_index)
        self.confmat += _multiclass_confusion_matrix_update(preds, target, self.num_classes)

    def compute(self) -> Tensor:
        return _jaccard_index_reduce(self.confmat, self.average, ignore_index=self.ignore_index, zero_division=self.zero_division)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MultilabelJaccardIndex(Metric):
    """Jaccard index for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0
    plot_legend_name: str = "Label"
    confmat: Tensor

    def __init__(
        self,
        num_labels: int,
        threshold: float = 0.5,
        average: Optional[str] = "macro",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        zero_division: float = 0.0,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.zero_division = zero_division
        if validate_args:
            _multilabel_confusion_matrix_arg_validation(num_labels, threshold, ignore_index, normalize=None)
        self.num_labels = num_labels
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.average = average

        self.add_state("confmat", torch.zeros(num_labels, 2, 2, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrices."""
        if self.validate_args:
            _multilabel_confusion_matrix_tensor_validation(preds, target, self.num_labels, self.ignore_index)
        preds, target = _multilabel_confusion_matrix_format(
            preds, target, self.num_labels, self.threshold, self.ignore_index
        )
        self.confmat += _multilabel_confusion_matrix_update(preds, target, self.num_labels)

    def compute(self) -> Tensor:
        return _jaccard_index_reduce(self.confmat, self.average, zero_division=self.zero_division)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class JaccardIndex(_ClassificationTaskWrapper):
    """Task-dispatching Jaccard index."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        threshold: float = 0.5,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        average: Optional[str] = "macro",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        kwargs.update({"ignore_index": ignore_index, "validate_args": validate_args})
        if task == ClassificationTask.BINARY:
            return BinaryJaccardIndex(threshold, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassJaccardIndex(num_classes, average, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelJaccardIndex(num_labels, threshold, average, **kwargs)
        raise ValueError(f"Not handled value: {task}")
