"""Modular stat-scores metrics.

Parity: torchmetrics ``classification/stat_scores.py`` — the
``_AbstractStatScores`` state container every count-based classification
metric builds on: tp/fp/tn/fn states are (C,)-long tensors with
``dist_reduce_fx='sum'`` (=> fused RCCL all-reduce at sync) for global
averaging, or ``cat`` list states for samplewise.
"""
from __future__ import annotations

from typing import Any, List, Optional, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.classification.base import _ClassificationTaskWrapper
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.utilities.enums import ClassificationTask
from metrics_amd.functional.classification.stat_scores import (
    _binary_stat_scores_arg_validation,
    _binary_stat_scores_compute,
    _binary_stat_scores_pipeline,
    _binary_stat_scores_tensor_validation,
    _multiclass_stat_scores_arg_validation,
    _multiclass_stat_scores_compute,
    _multiclass_stat_scores_pipeline,
    _multiclass_stat_scores_tensor_validation,
    _multilabel_stat_scores_arg_validation,
    _multilabel_stat_scores_compute,
    _multilabel_stat_scores_pipeline,
    _multilabel_stat_scores_tensor_validation,
)


class _AbstractStatScores(Metric):
    """Holds the tp/fp/tn/fn accumulators and the update/final-state plumbing."""

    tp: Union[List[Tensor], Tensor]
    fp: Union[List[Tensor], Tensor]
    tn: Union[List[Tensor], Tensor]
    fn: Union[List[Tensor], Tensor]

    def _create_state(self, size: int, multidim_average: str = "global") -> None:
        """Initialize the states for the different statistics."""
        default: Union[Tensor, List]
        if multidim_average == "samplewise":
            default = []
            dist_reduce_fx = "cat"
        else:
            default = torch.zeros(size, dtype=torch.long)
            dist_reduce_fx = "sum"

        def _default() -> Union[Tensor, List]:
            return default.detach().clone() if isinstance(default, Tensor) else []

        self.add_state("tp", _default(), dist_reduce_fx=dist_reduce_fx)
        self.add_state("fp", _default(), dist_reduce_fx=dist_reduce_fx)
        self.add_state("tn", _default(), dist_reduce_fx=dist_reduce_fx)
        self.add_state("fn", _default(), dist_reduce_fx=dist_reduce_fx)

    def _update_state(self, tp: Tensor, fp: Tensor, tn: Tensor, fn: Tensor) -> None:
        """Update states depending on multidim_average argument."""
        if self.multidim_average == "samplewise":
            self.tp.append(tp)
            self.fp.append(fp)
            self.tn.append(tn)
            self.fn.append(fn)
        else:
            self.tp += tp
            self.fp += fp
            self.tn += tn
            self.fn += fn

    def _final_state(self) -> tuple:
        """Aggregate states that are lists and return final states."""
        tp = dim_zero_cat(self.tp)
        fp = dim_zero_cat(self.fp)
        tn = dim_zero_cat(self.tn)
        fn = dim_zero_cat(self.fn)
        return tp, fp, tn, fn


class BinaryStatScores(_AbstractStatScores):
    """tp/fp/tn/fn counts for binary tasks."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False

    def __init__(
        self,
        threshold: float = 0.5,
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        zero_division = kwargs.pop("zero_division", 0)
        super(_AbstractStatScores, self).__init__(**kwargs)
        if validate_args:
            _binary_stat_scores_arg_validation(threshold, multidim_average, ignore_index, zero_division)
        self.threshold = threshold
        self.multidim_average = multidim_average
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.zero_division = zero_division

        self._create_state(size=1, multidim_average=multidim_average)

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate batch statistics (GPU: one fused HIP pass)."""
        if self.validate_args:
            _binary_stat_scores_tensor_validation(preds, target, self.multidim_average, self.ignore_index)
        tp, fp, tn, fn = _binary_stat_scores_pipeline(
            preds, target, self.threshold, self.multidim_average, self.ignore_index
        )
        self._update_state(tp, fp, tn, fn)

    def compute(self) -> Tensor:
        """Final statistics: (5,) tensor [tp, fp, tn, fn, support]."""
        tp, fp, tn, fn = self._final_state()
        return _binary_stat_scores_compute(tp, fp, tn, fn, self.multidim_average)


class MulticlassStatScores(_AbstractStatScores):
    _hip_fused_kind = "mc_stat"
    """Per-class tp/fp/tn/fn counts for multiclass tasks."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False

    def __init__(
        self,
        num_classes: Optional[int] = None,
        top_k: int = 1,
        average: Optional[str] = "macro",
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        zero_division = kwargs.pop("zero_division", 0)
        super(_AbstractStatScores, self).__init__(**kwargs)
        if num_classes is None and average != "micro":
            # reference semantics (functional/classification/stat_scores.py:238)
            raise ValueError(
                f"Argument `num_classes` can only be `None` for `average='micro'`, but got `average={average}`."
            )
        if validate_args and num_classes is not None:
            _multiclass_stat_scores_arg_validation(
                num_classes, top_k, average, multidim_average, ignore_index, zero_division
            )
        self.num_classes = num_classes
        self.top_k = top_k
        self.average = average
        self.multidim_average = multidim_average
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.zero_division = zero_division

        # per-class states regardless of averaging: micro reduces at compute
        # time, so micro- and macro-averaged metrics land in the SAME compute
        # group (one fused kernel launch per collection update). With
        # num_classes=None (micro-only) the class count is inferred per batch
        # and the totals land in size-1 states.
        self._create_state(size=num_classes if num_classes is not None else 1,
                           multidim_average=multidim_average)

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate batch statistics (GPU: fused argmax+count HIP kernel
        accumulating DIRECTLY into the tp/fp/tn/fn states — 3 launches)."""
        if self.num_classes is None:
            # micro-only convenience path: infer the class count per batch
            # and accumulate global totals into the size-1 states
            if preds.is_floating_point() and preds.ndim == target.ndim + 1:
                n_cls = preds.shape[1]
            else:
                n_cls = int(torch.maximum(preds.max(), target.max()).item()) + 1
            tp, fp, tn, fn = _multiclass_stat_scores_pipeline(
                preds, target, n_cls, self.top_k, "micro", self.multidim_average, self.ignore_index
            )
            self.tp = self.tp + tp.sum()
            self.fp = self.fp + fp.sum()
            self.tn = self.tn + tn.sum()
            self.fn = self.fn + fn.sum()
            return
        if self.validate_args:
            _multiclass_stat_scores_tensor_validation(
                preds, target, self.num_classes, self.multidim_average, self.ignore_index
            )
        if (
            preds.is_cuda
            and self.top_k == 1
            and self.multidim_average == "global"
            and isinstance(self.tp, Tensor)
            and (not preds.is_floating_point() or (preds.dtype in (torch.float32, torch.bfloat16)))
        ):
            from metrics_amd.ops import _hip

            if preds.ndim == target.ndim + 1 and preds.is_floating_point():
                p2 = preds.reshape(preds.shape[0], preds.shape[1], -1).movedim(1, -1).reshape(-1, preds.shape[1])
            else:
                p2 = preds.reshape(-1)
            scratch = getattr(self, "_hip_scratch", None)
            if scratch is None or scratch.device != preds.device:
                scratch = torch.zeros(3 * self.num_classes + 1, dtype=torch.long, device=preds.device)
                self._hip_scratch = scratch
            _hip.mc_stat_into(
                p2, target.reshape(-1), self.num_classes, self.ignore_index, scratch,
                self.tp, self.fp, self.tn, self.fn,
            )
            return
        tp, fp, tn, fn = _multiclass_stat_scores_pipeline(
            preds, target, self.num_classes, self.top_k, self.average, self.multidim_average, self.ignore_index
        )
        self._update_state(tp, fp, tn, fn)

    def compute(self) -> Tensor:
        """Final statistics, averaged per ``average``."""
        tp, fp, tn, fn = self._final_state()
        return _multiclass_stat_scores_compute(tp, fp, tn, fn, self.average, self.multidim_average)


class MultilabelStatScores(_AbstractStatScores):
    """Per-label tp/fp/tn/fn counts for multilabel tasks."""

    is_differentiable: bool = False
    higher_is_better: Optional[bool] = None
    full_state_update: bool = False

    def __init__(
        self,
        num_labels: int,
        threshold: float = 0.5,
        average: Optional[str] = "macro",
        multidim_average: str = "global",
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        zero_division = kwargs.pop("zero_division", 0)
        super(_AbstractStatScores, self).__init__(**kwargs)
        if validate_args:
            _multilabel_stat_scores_arg_validation(
                num_labels, threshold, average, multidim_average, ignore_index, zero_division
            )
        self.num_labels = num_labels
        self.threshold = threshold
        self.average = average
        self.multidim_average = multidim_average
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self.zero_division = zero_division

        self._create_state(size=num_labels, multidim_average=multidim_average)

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate batch statistics (GPU: one fused HIP pass)."""
        if self.validate_args:
            _multilabel_stat_scores_tensor_validation(
                preds, target, self.num_labels, self.multidim_average, self.ignore_index
            )
        tp, fp, tn, fn = _multilabel_stat_scores_pipeline(
            preds, target, self.num_labels, self.threshold, self.multidim_average, self.ignore_index
        )
        self._update_state(tp, fp, tn, fn)

    def compute(self) -> Tensor:
        """Final statistics, averaged per ``average``."""
        tp, fp, tn, fn = self._final_state()
        return _multilabel_stat_scores_compute(tp, fp, tn, fn, self.average, self.multidim_average)


class StatScores(_ClassificationTaskWrapper):
    """Task-dispatching StatScores: returns the Binary/Multiclass/Multilabel class."""

    def __new__(  # type: ignore[misc]
        cls,
        task: str,
        threshold: float = 0.5,
        num_classes: Optional[int] = None,
        num_labels: Optional[int] = None,
        average: Optional[str] = "micro",
        multidim_average: str = "global",
        top_k: Optional[int] = 1,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> Metric:
        task = ClassificationTask.from_str(task)
        assert multidim_average is not None  # noqa: S101
        kwargs.update({
            "multidim_average": multidim_average,
            "ignore_index": ignore_index,
            "validate_args": validate_args,
        })
        if task == ClassificationTask.BINARY:
            return BinaryStatScores(threshold, **kwargs)
        if task == ClassificationTask.MULTICLASS:
            if not isinstance(num_classes, int):
                raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
            if not isinstance(top_k, int):
                raise ValueError(f"`top_k` is expected to be `int` but `{type(top_k)} was passed.`")
            return MulticlassStatScores(num_classes, top_k, average, **kwargs)
        if task == ClassificationTask.MULTILABEL:
            if not isinstance(num_labels, int):
                raise ValueError(f"`num_labels` is expected to be `int` but `{type(num_labels)} was passed.`")
            return MultilabelStatScores(num_labels, threshold, average, **kwargs)
        raise ValueError(f"Not handled value: {task}")
