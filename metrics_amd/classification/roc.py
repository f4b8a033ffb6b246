"""Modular ROC. Parity: torchmetrics ``classification/roc.py``."""
from __future__ import annotations

from typing import Any, List, Optional, Tuple, Union

from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.classification.precision_recall_curve import (
    BinaryPrecisionRecallCurve,
    MulticlassPrecisionRecallCurve,
    MultilabelPrecisionRecallCurve,
)
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.roc import (
    _binary_roc_compute,
    _multiclass_roc_compute,
    _multilabel_roc_compute,
)


class BinaryROC(BinaryPrecisionRecallCurve):
    """ROC for binary tasks (stateful)."""

    def compute(self) -> Tuple[Tensor, Tensor, Tensor]:
        """(fpr, tpr, thresholds)."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _binary_roc_compute(state, self.thresholds)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.compute import _auc_compute_without_check
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        score = (
            _auc_compute_without_check(curve_computed[0], curve_computed[1], 1.0)
            if score is True
            else None
            if score is False
            else score
        )
        return plot_curve(
            curve_computed, score=score, ax=ax,
            label_names=("False positive rate", "True positive rate"), name=self.__class__.__name__,
        )


class MulticlassROC(MulticlassPrecisionRecallCurve):
    """ROC for multiclass tasks (stateful)."""

    def compute(self) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
        """(fpr, tpr, thresholds) per class."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _multiclass_roc_compute(state, self.num_classes, self.thresholds, self.average)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        return plot_curve(
            curve_computed, score=score if score is not True else None, ax=ax,
            label_names=("False positive rate", "True positive rate"), name=self.__class__.__name__,
        )


class MultilabelROC(MultilabelPrecisionRecallCurve):
    """ROC for multilabel tasks (stateful)."""

    def compute(self) -> Union[Tuple[Tensor, Tensor, Tensor], Tuple[List[Tensor], List[Tensor], List[Tensor]]]:
        """(fpr, tpr, thresholds) per label."""
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        return _multilabel_roc_compute(state, self.num_labels, self.thresholds, self.ignore_index)

    def plot(self, curve=None, score=None, ax=None):
        from metrics_amd.utilities.plot import plot_curve

        curve_computed = curve or self.compute()
        return plot_curve(
            curve_computed, score=score if score is not True else None, ax=ax,
            label_names=("False positive rate", "True positive rate"), name=self.__class__.__name__,
        )


class ROC(_ClassificationTaskWrapper):
    """Task-dispatching ROC."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        thresholds: Optional[Union[int, List[float], Tensor]] = None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        kwargs.update({"thresholds": thresholds, "ignore_index": ignore_index, "validate_args": validate_args})
        if task == ClassificationTask.BINARY:
            return BinaryROC(**kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassROC(num_classes, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelROC(num_labels, **kwargs)
        raise ValueError(f"Not handled value: {task}")
