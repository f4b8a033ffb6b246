"""Modular group fairness. Parity: torchmetrics ``classification/group_fairness.py``."""
from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.classification.group_fairness import (
    _binary_groups_stat_scores,
    _compute_binary_demographic_parity,
    _compute_binary_equal_opportunity,
    _groups_validation,
)
from metrics_amd.utilities.compute import _safe_divide


class _AbstractGroupStatScores(Metric):
    """Per-group tp/fp/tn/fn accumulators."""

    tp: Tensor
    fp: Tensor
    tn: Tensor
    fn: Tensor

    def _create_states(self, num_groups: int) -> None:
        default = lambda: torch.zeros(num_groups, dtype=torch.long)  # noqa: E731
        self.add_state("tp", default(), dist_reduce_fx="sum")
        self.add_state("fp", default(), dist_reduce_fx="sum")
        self.add_state("tn", default(), dist_reduce_fx="sum")
        self.add_state("fn", default(), dist_reduce_fx="sum")

    def _update_states(self, group_stats: List) -> None:
        for group, stats in enumerate(group_stats):
            tp, fp, tn, fn = stats
            self.tp[group] += tp
            self.fp[group] += fp
            self.tn[group] += tn
            self.fn[group] += fn


class BinaryGroupStatRates(_AbstractGroupStatScores):
    """Per-group tp/fp/tn/fn rates for fairness analysis (stateful)."""

    is_differentiable = False
    higher_is_better = False
    full_state_update = False

    def __init__(
        self,
        num_groups: int,
        threshold: float = 0.5,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if not isinstance(num_groups, int) and num_groups < 2:
            raise ValueError(f"Expected argument `num_groups` to be an int larger than 1, but got {num_groups}")
        self.num_groups = num_groups
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self._create_states(num_groups)

    def update(self, preds: Tensor, target: Tensor, groups: Tensor) -> None:
        """Accumulate per-group counts."""
        group_stats = _binary_groups_stat_scores(
            preds, target, groups, self.num_groups, self.threshold, self.ignore_index, self.validate_args
        )
        self._update_states(group_stats)

    def compute(self) -> Dict[str, Tensor]:
        """Per-group normalized [tp, fp, tn, fn]."""
        results = torch.stack([self.tp, self.fp, self.tn, self.fn], dim=1).float()
        results = results / results.sum(dim=1, keepdim=True)
        return {f"group_{i}": results[i] for i in range(self.num_groups)}


class BinaryFairness(_AbstractGroupStatScores):
    """Demographic parity / equal opportunity ratios between groups (stateful)."""

    is_differentiable = False
    higher_is_better = False
    full_state_update = False

    def __init__(
        self,
        num_groups: int,
        task: str = "all",
        threshold: float = 0.5,
        ignore_index: Optional[int] = None,
        validate_args: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if task not in ("demographic_parity", "equal_opportunity", "all"):
            raise ValueError(
                f"Expected argument `task` to either be 'demographic_parity', 'equal_opportunity' or 'all' but got {task}."
            )
        if not isinstance(num_groups, int) and num_groups < 2:
            raise ValueError(f"Expected argument `num_groups` to be an int larger than 1, but got {num_groups}")
        self.task = task
        self.num_groups = num_groups
        self.threshold = threshold
        self.ignore_index = ignore_index
        self.validate_args = validate_args
        self._create_states(num_groups)

    def update(self, preds: Tensor, target: Optional[Tensor] = None, groups: Optional[Tensor] = None) -> None:
        """Accumulate per-group counts (target unused for pure demographic parity)."""
        if groups is None:
            raise ValueError("Expected argument `groups` to be provided")
        if self.task == "demographic_parity":
            if target is not None:
                import warnings

                warnings.warn("The task demographic_parity does not require a target.", UserWarning, stacklevel=2)
            target = torch.zeros(preds.shape, dtype=torch.long, device=preds.device)
        group_stats = _binary_groups_stat_scores(
            preds, target, groups, self.num_groups, self.threshold, self.ignore_index, self.validate_args
        )
        self._update_states(group_stats)

    def compute(self) -> Dict[str, Tensor]:
        """min/max-group ratio for the configured fairness criteria."""
        out: Dict[str, Tensor] = {}
        if self.task in ("demographic_parity", "all"):
            rates = _compute_binary_demographic_parity(self.tp, self.fp, self.tn, self.fn)
            min_g = int(torch.argmin(rates))
            max_g = int(torch.argmax(rates))
            out[f"DP_{min_g}_{max_g}"] = _safe_divide(rates[min_g], rates[max_g])
        if self.task in ("equal_opportunity", "all"):
            rates = _compute_binary_equal_opportunity(self.tp, self.fp, self.tn, self.fn)
            min_g = int(torch.argmin(rates))
            max_g = int(torch.argmax(rates))
            out[f"EO_{min_g}_{max_g}"] = _safe_divide(rates[min_g], rates[max_g])
        return out


def _plot_group_stats(self, val=None, ax=None):
    """Plot the per-group rate dict."""
    return self._plot(val, ax)


BinaryGroupStatRates.plot = _plot_group_stats
BinaryFairness.plot = _plot_group_stats
