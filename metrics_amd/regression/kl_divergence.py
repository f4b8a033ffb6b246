"""Modular KLDivergence. Parity: torchmetrics ``regression/kl_divergence.py``."""
from __future__ import annotations

from typing import Any, List, Optional, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.functional.regression.kl_divergence import _kld_compute, _kld_update


class KLDivergence(Metric):
    """KL divergence (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update: bool = False
    plot_lower_bound: float = 0.0

    measures: Union[Tensor, List[Tensor]]
    total: Tensor

    def __init__(self, log_prob: bool = False, reduction: Optional[str] = "mean", **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(log_prob, bool):
            raise TypeError(f"Expected argument `log_prob` to be bool but got {log_prob}")
        self.log_prob = log_prob

        allowed_reduction = ("mean", "sum", "none", None)
        if reduction not in allowed_reduction:
            raise ValueError(f"Expected argument `reduction` to be one of {allowed_reduction} but got {reduction}")
        self.reduction = reduction

        if self.reduction in ("mean", "sum"):
            self.add_state("measures", torch.tensor(0.0), dist_reduce_fx="sum")
        else:
            self.add_state("measures", [], dist_reduce_fx="cat")
        self.add_state("total", torch.tensor(0), dist_reduce_fx="sum")

    def update(self, p: Tensor, q: Tensor) -> None:
        """Accumulate per-row KL terms."""
        measures, total = _kld_update(p, q, self.log_prob)
        if self.reduction is None or self.reduction == "none":
            self.measures.append(measures)
        else:
            self.measures = self.measures + measures.sum()
        self.total = self.total + total

    def compute(self) -> Tensor:
        """Reduced KL divergence."""
        measures = dim_zero_cat(self.measures) if self.reduction in ("none", None) else self.measures
        return _kld_compute(measures, self.total, self.reduction)

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)
