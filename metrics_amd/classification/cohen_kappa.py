"""Modular Cohen's kappa. Parity: torchmetrics ``classification/cohen_kappa.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.utilities.enums import ClassificationTaskNoMultilabel
from metrics_amd.functional.classification.cohen_kappa import _cohen_kappa_arg_validation, _cohen_kappa_reduce
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


class BinaryCohenKappa(Metric):
    """Cohen's kappa for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0
    confmat: Tensor

    def __init__(
        self,
        threshold: float = 0.5,
        weights: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _binary_confusion_matrix_arg_validation(threshold, ignore_index, normalize=None)
            _cohen_kappa_arg_validation(weights)
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.weights = weights

        self.add_state("confmat", torch.zeros(2, 2, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrix."""
        if self.validate_args:
            _binary_confusion_matrix_tensor_validation(preds, target, self.ignore_index)
        preds, target = _binary_confusion_matrix_format(preds, target, self.threshold, self.ignore_index)
        self.confmat += _binary_confusion_matrix_update(preds, target)

    def compute(self) -> Tensor:
        return _cohen_kappa_reduce(self.confmat, self.weights)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MulticlassCohenKappa(Metric):
    _hip_fused_kind = "mc_confmat"
    """Cohen's kappa for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0
    confmat: Tensor

    def __init__(
        self,
        num_classes: int,
        weights: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _multiclass_confusion_matrix_arg_validation(num_classes, ignore_index, normalize=None)
            _cohen_kappa_arg_validation(weights)
        self.num_classes = num_classes
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.weights = weights

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
        return _cohen_kappa_reduce(self.confmat, self.weights)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class CohenKappa(_ClassificationTaskWrapper):
    """Task-dispatching Cohen's kappa."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        threshold: float = 0.5,
        num_classes: Optional[int] = None,
        weights: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTaskNoMultilabel.from_str(task)
        kwargs.update({"weights": weights, "ignore_index": ignore_index, "validate_args": validate_args})
        if task == ClassificationTaskNoMultilabel.BINARY:
            return BinaryCohenKappa(threshold, **kwargs)
        if task == ClassificationTaskNoMultilabel.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassCohenKappa(num_classes, **kwargs)
        raise ValueError(f"Not handled value: {task}")
