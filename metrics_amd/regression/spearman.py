"""Modular Spearman correlation. Parity: torchmetrics ``regression/spearman.py``."""
from __future__ import annotations

from typing import Any, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.functional.regression.spearman import _spearman_corrcoef_compute, _spearman_corrcoef_update


class SpearmanCorrCoef(Metric):
    """Spearman rank correlation (stateful; cat state, rank at compute)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = -1.0
    plot_upper_bound: float = 1.0

    preds: List[Tensor]
    target: List[Tensor]

    def __init__(self, num_outputs: int = 1, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(num_outputs, int) and num_outputs < 1:
            raise ValueError("Expected argument `num_outputs` to be an int larger than 0, but got {num_outputs}")
        self.num_outputs = num_outputs
        self.add_state("preds", default=[], dist_reduce_fx="cat")
        self.add_state("target", default=[], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Append the batch."""
        preds, target = _spearman_corrcoef_update(preds, target, self.num_outputs)
        self.preds.append(preds)
        self.target.append(target)

    def compute(self) -> Tensor:
        """Spearman correlation over all accumulated data."""
        preds = dim_zero_cat(self.preds)
        target = dim_zero_cat(self.target)
        return _spearman_corrcoef_compute(preds, target)

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)


class KendallRankCorrCoef(Metric):
    """Kendall rank correlation (stateful; cat state)."""

    is_differentiable = False
    higher_is_better = None
    full_state_update: bool = False
    plot_lower_bound: float = -1.0
    plot_upper_bound: float = 1.0

    preds: List[Tensor]
    target: List[Tensor]

    def __init__(
        self,
        variant: str = "b",
        t_test: bool = False,
        alternative: Optional[str] = "two-sided",
        num_outputs: int = 1,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if variant not in ("a", "b", "c"):
            raise ValueError(f"Argument `variant` is expected to be one of 'a', 'b', 'c', got {variant}")
        if not isinstance(t_test, bool):
            raise ValueError(f"Argument `t_test` is expected to be of a type `bool`, but got {t_test}.")
        if t_test and alternative is None:
            raise ValueError("Argument `alternative` is required if `t_test=True` but got `None`.")
        self.variant = variant
        self.alternative = alternative if t_test else None
        self.t_test = t_test
        self.num_outputs = num_outputs
        self.add_state("preds", default=[], dist_reduce_fx="cat")
        self.add_state("target", default=[], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Append the batch."""
        self.preds.append(preds)
        self.target.append(target)

    def compute(self):
        """Kendall tau (and p-value when t_test)."""
        from metrics_amd.functional.regression.kendall import _kendall_corrcoef_compute

        preds = dim_zero_cat(self.preds)
        target = dim_zero_cat(self.target)
        tau, p = _kendall_corrcoef_compute(preds, target, self.variant, self.alternative)
        if self.t_test:
            return tau, p
        return tau

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)
