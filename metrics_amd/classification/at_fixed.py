"""Modular at-fixed-X / LogAUC metrics. Parity: torchmetrics
``classification/{recall_fixed_precision,precision_fixed_recall,
sensitivity_specificity,specificity_sensitivity,logauc}.py``."""
from __future__ import annotations

from typing import Any, List, Optional, Tuple, Union

import torch
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
from metrics_amd.functional.classification.at_fixed import (
    _binary_roc_compute,
    _binary_precision_recall_curve_compute,
    _logauc_compute,
    _multiclass_precision_recall_curve_compute,
    _multiclass_roc_compute,
    _multilabel_precision_recall_curve_compute,
    _multilabel_roc_compute,
    _precision_at_recall,
    _recall_at_precision,
    _sens_at_spec,
    _spec_at_sens,
)


class BinaryRecallAtFixedPrecision(BinaryPrecisionRecallCurve):
    """Max recall at fixed precision for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, min_precision: float, thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(thresholds, ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_precision, float) or not (0 <= min_precision <= 1)):
            raise ValueError(f"Expected argument `min_precision` to be a float in the [0,1] range, but got {min_precision}")
        self.validate_args = validate_args
        self.min_precision = min_precision

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _binary_precision_recall_curve_compute(state, self.thresholds)
        return _recall_at_precision(a, b, th, self.min_precision)


class MulticlassRecallAtFixedPrecision(MulticlassPrecisionRecallCurve):
    """Max recall at fixed precision for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_classes: int, min_precision: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_classes, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_precision, float) or not (0 <= min_precision <= 1)):
            raise ValueError(f"Expected argument `min_precision` to be a float in the [0,1] range, but got {min_precision}")
        self.validate_args = validate_args
        self.min_precision = min_precision

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multiclass_precision_recall_curve_compute(state, self.num_classes, self.thresholds, None)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_recall_at_precision(a[i], b[i], th, self.min_precision) for i in range(self.num_classes)]
        else:
            res = [_recall_at_precision(a[i], b[i], th[i], self.min_precision) for i in range(self.num_classes)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class MultilabelRecallAtFixedPrecision(MultilabelPrecisionRecallCurve):
    """Max recall at fixed precision for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_labels: int, min_precision: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_labels, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_precision, float) or not (0 <= min_precision <= 1)):
            raise ValueError(f"Expected argument `min_precision` to be a float in the [0,1] range, but got {min_precision}")
        self.validate_args = validate_args
        self.min_precision = min_precision

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multilabel_precision_recall_curve_compute(state, self.num_labels, self.thresholds, self.ignore_index)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_recall_at_precision(a[i], b[i], th, self.min_precision) for i in range(self.num_labels)]
        else:
            res = [_recall_at_precision(a[i], b[i], th[i], self.min_precision) for i in range(self.num_labels)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class RecallAtFixedPrecision(_ClassificationTaskWrapper):
    """Task-dispatching Max recall at fixed precision."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        min_precision: float,
        thresholds=None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        if task == ClassificationTask.BINARY:
            return BinaryRecallAtFixedPrecision(min_precision, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassRecallAtFixedPrecision(num_classes, min_precision, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelRecallAtFixedPrecision(num_labels, min_precision, thresholds, ignore_index, validate_args, **kwargs)
        raise ValueError(f"Not handled value: {task}")


class BinaryPrecisionAtFixedRecall(BinaryPrecisionRecallCurve):
    """Max precision at fixed recall for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, min_recall: float, thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(thresholds, ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_recall, float) or not (0 <= min_recall <= 1)):
            raise ValueError(f"Expected argument `min_recall` to be a float in the [0,1] range, but got {min_recall}")
        self.validate_args = validate_args
        self.min_recall = min_recall

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _binary_precision_recall_curve_compute(state, self.thresholds)
        return _precision_at_recall(a, b, th, self.min_recall)


class MulticlassPrecisionAtFixedRecall(MulticlassPrecisionRecallCurve):
    """Max precision at fixed recall for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_classes: int, min_recall: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_classes, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_recall, float) or not (0 <= min_recall <= 1)):
            raise ValueError(f"Expected argument `min_recall` to be a float in the [0,1] range, but got {min_recall}")
        self.validate_args = validate_args
        self.min_recall = min_recall

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multiclass_precision_recall_curve_compute(state, self.num_classes, self.thresholds, None)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_precision_at_recall(a[i], b[i], th, self.min_recall) for i in range(self.num_classes)]
        else:
            res = [_precision_at_recall(a[i], b[i], th[i], self.min_recall) for i in range(self.num_classes)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class MultilabelPrecisionAtFixedRecall(MultilabelPrecisionRecallCurve):
    """Max precision at fixed recall for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_labels: int, min_recall: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_labels, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_recall, float) or not (0 <= min_recall <= 1)):
            raise ValueError(f"Expected argument `min_recall` to be a float in the [0,1] range, but got {min_recall}")
        self.validate_args = validate_args
        self.min_recall = min_recall

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multilabel_precision_recall_curve_compute(state, self.num_labels, self.thresholds, self.ignore_index)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_precision_at_recall(a[i], b[i], th, self.min_recall) for i in range(self.num_labels)]
        else:
            res = [_precision_at_recall(a[i], b[i], th[i], self.min_recall) for i in range(self.num_labels)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class PrecisionAtFixedRecall(_ClassificationTaskWrapper):
    """Task-dispatching Max precision at fixed recall."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        min_recall: float,
        thresholds=None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        if task == ClassificationTask.BINARY:
            return BinaryPrecisionAtFixedRecall(min_recall, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassPrecisionAtFixedRecall(num_classes, min_recall, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelPrecisionAtFixedRecall(num_labels, min_recall, thresholds, ignore_index, validate_args, **kwargs)
        raise ValueError(f"Not handled value: {task}")


class BinarySensitivityAtSpecificity(BinaryPrecisionRecallCurve):
    """Max sensitivity at fixed specificity for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, min_specificity: float, thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(thresholds, ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_specificity, float) or not (0 <= min_specificity <= 1)):
            raise ValueError(f"Expected argument `min_specificity` to be a float in the [0,1] range, but got {min_specificity}")
        self.validate_args = validate_args
        self.min_specificity = min_specificity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _binary_roc_compute(state, self.thresholds)
        return _sens_at_spec(a, b, th, self.min_specificity)


class MulticlassSensitivityAtSpecificity(MulticlassPrecisionRecallCurve):
    """Max sensitivity at fixed specificity for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_classes: int, min_specificity: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_classes, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_specificity, float) or not (0 <= min_specificity <= 1)):
            raise ValueError(f"Expected argument `min_specificity` to be a float in the [0,1] range, but got {min_specificity}")
        self.validate_args = validate_args
        self.min_specificity = min_specificity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multiclass_roc_compute(state, self.num_classes, self.thresholds)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_sens_at_spec(a[i], b[i], th, self.min_specificity) for i in range(self.num_classes)]
        else:
            res = [_sens_at_spec(a[i], b[i], th[i], self.min_specificity) for i in range(self.num_classes)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class MultilabelSensitivityAtSpecificity(MultilabelPrecisionRecallCurve):
    """Max sensitivity at fixed specificity for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_labels: int, min_specificity: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_labels, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_specificity, float) or not (0 <= min_specificity <= 1)):
            raise ValueError(f"Expected argument `min_specificity` to be a float in the [0,1] range, but got {min_specificity}")
        self.validate_args = validate_args
        self.min_specificity = min_specificity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multilabel_roc_compute(state, self.num_labels, self.thresholds, self.ignore_index)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_sens_at_spec(a[i], b[i], th, self.min_specificity) for i in range(self.num_labels)]
        else:
            res = [_sens_at_spec(a[i], b[i], th[i], self.min_specificity) for i in range(self.num_labels)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class SensitivityAtSpecificity(_ClassificationTaskWrapper):
    """Task-dispatching Max sensitivity at fixed specificity."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        min_specificity: float,
        thresholds=None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        if task == ClassificationTask.BINARY:
            return BinarySensitivityAtSpecificity(min_specificity, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassSensitivityAtSpecificity(num_classes, min_specificity, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelSensitivityAtSpecificity(num_labels, min_specificity, thresholds, ignore_index, validate_args, **kwargs)
        raise ValueError(f"Not handled value: {task}")


class BinarySpecificityAtSensitivity(BinaryPrecisionRecallCurve):
    """Max specificity at fixed sensitivity for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, min_sensitivity: float, thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(thresholds, ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_sensitivity, float) or not (0 <= min_sensitivity <= 1)):
            raise ValueError(f"Expected argument `min_sensitivity` to be a float in the [0,1] range, but got {min_sensitivity}")
        self.validate_args = validate_args
        self.min_sensitivity = min_sensitivity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _binary_roc_compute(state, self.thresholds)
        return _spec_at_sens(a, b, th, self.min_sensitivity)


class MulticlassSpecificityAtSensitivity(MulticlassPrecisionRecallCurve):
    """Max specificity at fixed sensitivity for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_classes: int, min_sensitivity: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_classes, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_sensitivity, float) or not (0 <= min_sensitivity <= 1)):
            raise ValueError(f"Expected argument `min_sensitivity` to be a float in the [0,1] range, but got {min_sensitivity}")
        self.validate_args = validate_args
        self.min_sensitivity = min_sensitivity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multiclass_roc_compute(state, self.num_classes, self.thresholds)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_spec_at_sens(a[i], b[i], th, self.min_sensitivity) for i in range(self.num_classes)]
        else:
            res = [_spec_at_sens(a[i], b[i], th[i], self.min_sensitivity) for i in range(self.num_classes)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class MultilabelSpecificityAtSensitivity(MultilabelPrecisionRecallCurve):
    """Max specificity at fixed sensitivity for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_labels: int, min_sensitivity: float, thresholds=None, ignore_index=None,
                 validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_labels, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        if validate_args and (not isinstance(min_sensitivity, float) or not (0 <= min_sensitivity <= 1)):
            raise ValueError(f"Expected argument `min_sensitivity` to be a float in the [0,1] range, but got {min_sensitivity}")
        self.validate_args = validate_args
        self.min_sensitivity = min_sensitivity

    def compute(self) -> Tuple[Tensor, Tensor]:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        a, b, th = _multilabel_roc_compute(state, self.num_labels, self.thresholds, self.ignore_index)
        if isinstance(a, Tensor) and a.ndim == 2:
            res = [_spec_at_sens(a[i], b[i], th, self.min_sensitivity) for i in range(self.num_labels)]
        else:
            res = [_spec_at_sens(a[i], b[i], th[i], self.min_sensitivity) for i in range(self.num_labels)]
        return torch.stack([r[0] for r in res]), torch.stack([torch.as_tensor(r[1]) for r in res])


class SpecificityAtSensitivity(_ClassificationTaskWrapper):
    """Task-dispatching Max specificity at fixed sensitivity."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        min_sensitivity: float,
        thresholds=None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        if task == ClassificationTask.BINARY:
            return BinarySpecificityAtSensitivity(min_sensitivity, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            return MulticlassSpecificityAtSensitivity(num_classes, min_sensitivity, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelSpecificityAtSensitivity(num_labels, min_sensitivity, thresholds, ignore_index, validate_args, **kwargs)
        raise ValueError(f"Not handled value: {task}")


class BinaryLogAUC(BinaryPrecisionRecallCurve):
    """Log AUC for binary tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    def __init__(self, fpr_range: Tuple[float, float] = (0.001, 0.1), thresholds=None, ignore_index=None,
                 validate_args: bool = False, **kwargs: Any) -> None:
        super().__init__(thresholds, ignore_index, validate_args=False, **kwargs)
        self.fpr_range = tuple(fpr_range)
        self.validate_args = validate_args

    def compute(self) -> Tensor:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        fpr, tpr, _ = _binary_roc_compute(state, self.thresholds)
        return _logauc_compute(fpr, tpr, self.fpr_range)


class MulticlassLogAUC(MulticlassPrecisionRecallCurve):
    """Log AUC for multiclass tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_classes: int, fpr_range: Tuple[float, float] = (0.001, 0.1), average: Optional[str] = None,
                 thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_classes, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        self.fpr_range = tuple(fpr_range)
        self.average2 = average
        self.validate_args = validate_args

    def compute(self) -> Tensor:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        fpr, tpr, _ = _multiclass_roc_compute(state, self.num_classes, self.thresholds)
        scores = torch.stack([_logauc_compute(fpr[i], tpr[i], self.fpr_range) for i in range(self.num_classes)])
        if self.average2 == "macro":
            return scores.mean()
        return scores


class MultilabelLogAUC(MultilabelPrecisionRecallCurve):
    """Log AUC for multilabel tasks (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False

    def __init__(self, num_labels: int, fpr_range: Tuple[float, float] = (0.001, 0.1), average: Optional[str] = None,
                 thresholds=None, ignore_index=None, validate_args: bool = True, **kwargs: Any) -> None:
        super().__init__(num_labels, thresholds, ignore_index=ignore_index, validate_args=False, **kwargs)
        self.fpr_range = tuple(fpr_range)
        self.average2 = average
        self.validate_args = validate_args

    def compute(self) -> Tensor:
        state = (dim_zero_cat(self.preds), dim_zero_cat(self.target)) if self.thresholds is None else self.confmat
        fpr, tpr, _ = _multilabel_roc_compute(state, self.num_labels, self.thresholds, self.ignore_index)
        scores = torch.stack([_logauc_compute(fpr[i], tpr[i], self.fpr_range) for i in range(self.num_labels)])
        if self.average2 == "macro":
            return scores.mean()
        return scores


class LogAUC(_ClassificationTaskWrapper):
    """Task-dispatching LogAUC."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        fpr_range: Tuple[float, float] = (0.001, 0.1),
        thresholds=None,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        average: Optional[str] = "macro",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        if task == ClassificationTask.BINARY:
            return BinaryLogAUC(fpr_range, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            return MulticlassLogAUC(num_classes, fpr_range, average, thresholds, ignore_index, validate_args, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            return MultilabelLogAUC(num_labels, fpr_range, average, thresholds, ignore_index, validate_args, **kwargs)
        raise ValueError(f"Not handled value: {task}")


def _plot_first_value(self, val=None, ax=None):
    """Plot the metric value; defaults to the value component of (value, threshold)."""
    if val is None:
        val = self.compute()[0]
    return self._plot(val, ax)


def _plot_scalar(self, val=None, ax=None):
    """Plot the (scalar) metric value."""
    return self._plot(val, ax)


# the at-fixed metrics compute (value, threshold) tuples and LogAUC a scalar —
# neither can use the inherited precision-recall-curve plot
for _cls in (
    BinaryRecallAtFixedPrecision, MulticlassRecallAtFixedPrecision, MultilabelRecallAtFixedPrecision,
    BinaryPrecisionAtFixedRecall, MulticlassPrecisionAtFixedRecall, MultilabelPrecisionAtFixedRecall,
    BinarySensitivityAtSpecificity, MulticlassSensitivityAtSpecificity, MultilabelSensitivityAtSpecificity,
    BinarySpecificityAtSensitivity, MulticlassSpecificityAtSensitivity, MultilabelSpecificityAtSensitivity,
):
    _cls.plot = _plot_first_value
for _cls in (BinaryLogAUC, MulticlassLogAUC, MultilabelLogAUC):
    _cls.plot = _plot_scalar
