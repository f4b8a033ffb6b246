"""Modular ConfusionMatrix. Parity: torchmetrics ``classification/confusion_matrix.py``.

GPU multiclass update: ONE fused HIP kernel (argmax + confmat atomics) added
directly into the (C,C) state — no intermediates.
"""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd import ops
from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.confusion_matrix import (
    _binary_confusion_matrix_arg_validation,
    _binary_confusion_matrix_compute,
    _binary_confusion_matrix_format,
    _binary_confusion_matrix_tensor_validation,
    _binary_confusion_matrix_update,
    _multiclass_confusion_matrix_arg_validation,
    _multiclass_confusion_matrix_compute,
    _multiclass_confusion_matrix_format,
    _multiclass_confusion_matrix_tensor_validation,
    _multiclass_confusion_matrix_update,
    _multilabel_confusion_matrix_arg_validation,
    _multilabel_confusion_matrix_compute,
    _multilabel_confusion_matrix_format,
    _multilabel_confusion_matrix_tensor_validation,
    _multilabel_confusion_matrix_update,
)


class BinaryConfusionMatrix(Metric):
    """(2,2) confusion matrix for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = None
    full_state_update: bool = False
    confmat: Tensor

    def __init__(
        self,
        threshold: float = 0.5,
        ignore_index: Optional[int] = None,
        normalize: Optional[str] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _binary_confusion_matrix_arg_validation(threshold, ignore_index, normalize)
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.normalize = normalize
        self.validate_args = validate_args

        self.add_state("confmat", torch.zeros(2, 2, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrix."""
        if self.validate_args:
            _binary_confusion_matrix_tensor_validation(preds, target, self.ignore_index)
        preds, target = _binary_confusion_matrix_format(preds, target, self.threshold, self.ignore_index)
        confmat = _binary_confusion_matrix_update(preds, target)
        self.confmat += confmat

    def compute(self) -> Tensor:
        """Final (normalized) confusion matrix."""
        return _binary_confusion_matrix_compute(self.confmat, self.normalize)

    def plot(self, val: Optional[Tensor] = None, ax: Optional[Any] = None, add_text: bool = True, labels=None, cmap=None):
        from metrics_amd.utilities.plot import plot_confusion_matrix

        val = val if val is not None else self.compute()
        if not isinstance(val, Tensor):
            raise TypeError(f"Expected val to be a single tensor but got {val}")
        return plot_confusion_matrix(val, ax=ax, add_text=add_text, labels=labels, cmap=cmap)


def _fused_confmat_update(metric, preds, target) -> bool:
    """GPU fast path shared by every confmat-state metric (ConfusionMatrix,
    CohenKappa, MatthewsCorrCoef, JaccardIndex): accumulate the batch counts
    atomically into ``metric.confmat`` — no per-batch confmat, fill, or add.
    Returns False when the caller must take the torch path."""
    if not (preds.is_cuda and (not preds.is_floating_point() or preds.dtype in (torch.float32, torch.bfloat16))):
        return False
    if preds.ndim == target.ndim + 1 and preds.is_floating_point():
        p2 = preds.reshape(preds.shape[0], preds.shape[1], -1).movedim(1, -1).reshape(-1, preds.shape[1])
    else:
        p2 = preds.reshape(-1)
    from metrics_amd.ops import _hip

    dummy = getattr(metric, "_hip_dummy", None)
    if dummy is None or dummy.device != preds.device:
        dummy = torch.zeros(3 * metric.num_classes + 1, dtype=torch.long, device=preds.device)
        metric._hip_dummy = dummy
    _hip.mc_confmat_into(p2, target.reshape(-1), metric.num_classes, metric.ignore_index, metric.confmat, dummy)
    return True


class MulticlassConfusionMatrix(Metric):
    _hip_fused_kind = "mc_confmat"
    """(C,C) confusion matrix for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = None
    full_state_update: bool = False
    confmat: Tensor

    def __init__(
        self,
        num_classes: int,
        ignore_index: Optional[int] = None,
        normalize: Optional[str] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _multiclass_confusion_matrix_arg_validation(num_classes, ignore_index, normalize)
        self.num_classes = num_classes
        self.ignore_index = ignore_index
        self.normalize = normalize
        self.validate_args = validate_args

        self.add_state("confmat", torch.zeros(num_classes, num_classes, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrix (GPU: fused HIP kernel)."""
        if self.validate_args:
            _multiclass_confusion_matrix_tensor_validation(preds, target, self.num_classes, self.ignore_index)
        if _fused_confmat_update(self, preds, target):
            return
        preds, target = _multiclass_confusion_matrix_format(preds, target, self.ignore_index)
        confmat = _multiclass_confusion_matrix_update(preds, target, self.num_classes)
        self.confmat += confmat

    def compute(self) -> Tensor:
        """Final (normalized) confusion matrix."""
        return _multiclass_confusion_matrix_compute(self.confmat, self.normalize)

    def plot(self, val: Optional[Tensor] = None, ax: Optional[Any] = None, add_text: bool = True, labels=None, cmap=None):
        from metrics_amd.utilities.plot import plot_confusion_matrix

        val = val if val is not None else self.compute()
        if not isinstance(val, Tensor):
            raise TypeError(f"Expected val to be a single tensor but got {val}")
        return plot_confusion_matrix(val, ax=ax, add_text=add_text, labels=labels, cmap=cmap)


class MultilabelConfusionMatrix(Metric):
    """(L,2,2) confusion matrices for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = None
    full_state_update: bool = False
    confmat: Tensor

    def __init__(
        self,
        num_labels: int,
        threshold: float = 0.5,
        ignore_index: Optional[int] = None,
        normalize: Optional[str] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if validate_args:
            _multilabel_confusion_matrix_arg_validation(num_labels, threshold, ignore_index, normalize)
        self.num_labels = num_labels
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.normalize = normalize
        self.validate_args = validate_args

        self.add_state("confmat", torch.zeros(num_labels, 2, 2, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the batch confusion matrices."""
        if self.validate_args:
            _multilabel_confusion_matrix_tensor_validation(preds, target, self.num_labels, self.ignore_index)
        preds, target = _multilabel_confusion_matrix_format(
            preds, target, self.num_labels, self.threshold, self.ignore_index
        )
        confmat = _multilabel_confusion_matrix_update(preds, target, self.num_labels)
        self.confmat += confmat

    def compute(self) -> Tensor:
        """Final (normalized) confusion matrices."""
        return _multilabel_confusion_matrix_compute(self.confmat, self.normalize)

    def plot(self, val: Optional[Tensor] = None, ax: Optional[Any] = None, add_text: bool = True, labels=None, cmap=None):
        from metrics_amd.utilities.plot import plot_confusion_matrix

        val = val if val is not None else self.compute()
        if not isinstance(val, Tensor):
            raise TypeError(f"Expected val to be a single tensor but got {val}")
        return plot_confusion_matrix(val, ax=ax, add_text=add_text, labels=labels, cmap=cmap)


class ConfusionMatrix(_ClassificationTaskWrapper):
    """Task-dispatching ConfusionMatrix."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        threshold: float = 0.5,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        normalize: Optional[str] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        kwargs.update({"normalize": normalize, "ignore_index": ignore_index, "validate_args": validate_args})
        if task == ClassificationTask.BINARY:
            return BinaryConfusionMatrix(threshold, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassConfusionMatrix(num_classes, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelConfusionMatrix(num_labels, threshold, **kwargs)
        raise ValueError(f"Not handled value: {task}")
