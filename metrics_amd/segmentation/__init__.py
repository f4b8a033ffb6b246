"""Modular segmentation metrics. Parity: torchmetrics ``segmentation/*``."""
from __future__ import annotations

from typing import Any, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.functional.segmentation.metrics import (
    _dice_score_compute,
    _dice_score_update,
    _mean_iou_compute,
    _mean_iou_update,
    generalized_dice_score,
    hausdorff_distance,
)


class MeanIoU(Metric):
    """Mean IoU for semantic segmentation (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    score: Tensor
    num_batches: Tensor

    def __init__(
        self,
        num_classes: int,
        include_background: bool = True,
        per_class: bool = False,
        input_format: str = "one-hot",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if num_classes is not None and not (isinstance(num_classes, int) and num_classes > 0):
            raise ValueError(f"Expected argument `num_classes` must be `None` or a positive integer, but got {num_classes}")
        if not isinstance(include_background, bool):
            raise ValueError(f"Expected argument `include_background` must be a boolean, but got {include_background}")
        if not isinstance(per_class, bool):
            raise ValueError(f"Expected argument `per_class` must be a boolean, but got {per_class}")
        self.num_classes = num_classes
        self.include_background = include_background
        self.per_class = per_class
        self.input_format = input_format

        num_stat_classes = (num_classes - (0 if include_background else 1)) if num_classes else 1
        self.add_state("score", default=torch.zeros(num_stat_classes if per_class else 1).squeeze(), dist_reduce_fx="sum")
        self.add_state("num_batches", default=torch.tensor(0), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-class IoU sums."""
        from metrics_amd.functional.segmentation import mean_iou as _miou_fn

        score = _miou_fn(
            preds, target, self.num_classes, self.include_background, self.per_class, self.input_format
        )
        # reference segmentation/mean_iou.py:117-128: running mean over batches
        self.score += score.mean(0) if self.per_class else score.mean()
        self.num_batches += 1

    def compute(self) -> Tensor:
        """Average IoU over updates."""
        return self.score / self.num_batches

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class DiceScore(Metric):
    """Dice score for semantic segmentation (stateful; cat states)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    numerator: List[Tensor]
    denominator: List[Tensor]
    support: List[Tensor]

    def __init__(
        self,
        num_classes: int,
        include_background: bool = True,
        average: Optional[str] = "micro",
        input_format: str = "one-hot",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if average not in ("micro", "macro", "weighted", "none", None):
            raise ValueError(f"Expected argument `average` to be one of 'micro', 'macro', 'weighted', 'none', None, but got {average}")
        self.num_classes = num_classes
        self.include_background = include_background
        self.average = average
        self.input_format = input_format

        self.add_state("numerator", [], dist_reduce_fx="cat")
        self.add_state("denominator", [], dist_reduce_fx="cat")
        self.add_state("support", [], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample dice parts."""
        numerator, denominator, support = _dice_score_update(
            preds, target, self.num_classes, self.include_background, self.input_format
        )
        self.numerator.append(numerator)
        self.denominator.append(denominator)
        self.support.append(support)

    def compute(self) -> Tensor:
        """Dice averaged over samples."""
        numerator = dim_zero_cat(self.numerator)
        denominator = dim_zero_cat(self.denominator)
        support = dim_zero_cat(self.support)
        dice = _dice_score_compute(numerator, denominator, self.average, support)
        return dice.nanmean(0) if dice.ndim > 0 else dice

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class GeneralizedDiceScore(Metric):
    """Generalized dice score (stateful)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    score: Tensor
    samples: Tensor

    def __init__(
        self,
        num_classes: int,
        include_background: bool = True,
        per_class: bool = False,
        weight_type: str = "square",
        input_format: str = "one-hot",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.num_classes = num_classes
        self.include_background = include_background
        self.per_class = per_class
        self.weight_type = weight_type
        self.input_format = input_format

        num_stat_classes = num_classes - (0 if include_background else 1)
        self.add_state("score", default=torch.zeros(num_stat_classes if per_class else 1), dist_reduce_fx="sum")
        self.add_state("samples", default=torch.zeros(1, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate generalized dice sums."""
        score = generalized_dice_score(
            preds, target, self.num_classes, self.include_background, self.per_class,
            self.weight_type, self.input_format,
        )
        self.score += score.sum(0)
        self.samples += preds.shape[0]

    def compute(self) -> Tensor:
        """Average generalized dice."""
        return (self.score / self.samples).squeeze()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class HausdorffDistance(Metric):
    """Hausdorff distance (stateful)."""

    is_differentiable = False
    higher_is_better = False
    full_state_update: bool = False
    plot_lower_bound: float = 0.0

    score: Tensor
    total: Tensor

    def __init__(
        self,
        num_classes: int,
        include_background: bool = False,
        distance_metric: str = "euclidean",
        spacing=None,
        directed: bool = False,
        input_format: str = "one-hot",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if distance_metric not in ("euclidean", "chessboard", "taxicab"):
            raise ValueError(
                f"Argument `distance_metric` must be one of `euclidean`, `chessboard`, `taxicab`, but got {distance_metric}."
            )
        self.num_classes = num_classes
        self.include_background = include_background
        self.distance_metric = distance_metric
        self.spacing = spacing
        self.directed = directed
        self.input_format = input_format

        self.add_state("score", default=torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("total", default=torch.tensor(0), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate hausdorff distances."""
        hd = hausdorff_distance(
            preds, target, self.num_classes, self.include_background, self.distance_metric,
            self.spacing, self.directed, self.input_format,
        )
        self.score += hd.sum()
        self.total += hd.numel()

    def compute(self) -> Tensor:
        """Average hausdorff distance over all samples and classes."""
        return self.score / self.total

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


__all__ = ["DiceScore", "GeneralizedDiceScore", "HausdorffDistance", "MeanIoU"]
