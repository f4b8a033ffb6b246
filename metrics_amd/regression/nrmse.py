"""Modular NormalizedRootMeanSquaredError. Parity: torchmetrics ``regression/nrmse.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.regression.nrmse import _normalized_root_mean_squared_error_update


class NormalizedRootMeanSquaredError(Metric):
    """NRMSE (stateful). The denominator statistic is tracked across batches."""

    is_differentiable = True
    higher_is_better = False
    full_state_update: bool = False
    plot_lower_bound: float = 0.0

    sum_squared_error: Tensor
    total: Tensor

    def __init__(self, normalization: str = "mean", num_outputs: int = 1, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if normalization not in ("mean", "range", "std", "l2"):
            raise ValueError(
                f"Argument `normalization` should be either 'mean', 'range', 'std' or 'l2', but got {normalization}"
            )
        self.normalization = normalization
        if not (isinstance(num_outputs, int) and num_outputs > 0):
            raise ValueError(f"Expected num_outputs to be a positive integer but got {num_outputs}")
        self.num_outputs = num_outputs

        self.add_state("sum_squared_error", default=torch.zeros(num_outputs), dist_reduce_fx="sum")
        self.add_state("total", default=torch.tensor(0.0), dist_reduce_fx="sum")
        # streaming stats for the denominator
        self.add_state("target_sum", default=torch.zeros(num_outputs), dist_reduce_fx="sum")
        self.add_state("target_squared_sum", default=torch.zeros(num_outputs), dist_reduce_fx="sum")
        self.add_state("target_min", default=torch.full((num_outputs,), float("inf")), dist_reduce_fx="min")
        self.add_state("target_max", default=torch.full((num_outputs,), -float("inf")), dist_reduce_fx="max")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate squared error + target statistics."""
        from metrics_amd.functional.regression.mse import _mean_squared_error_update

        sum_squared_error, num_obs = _mean_squared_error_update(preds, target, self.num_outputs)
        self.sum_squared_error = self.sum_squared_error + sum_squared_error
        self.total = self.total + num_obs
        t = target.view(-1) if self.num_outputs == 1 else target
        self.target_sum = self.target_sum + t.sum(0)
        self.target_squared_sum = self.target_squared_sum + (t * t).sum(0)
        self.target_min = torch.minimum(self.target_min, t.min(0).values)
        self.target_max = torch.maximum(self.target_max, t.max(0).values)

    def compute(self) -> Tensor:
        """NRMSE with the configured normalization."""
        rmse = torch.sqrt(self.sum_squared_error / self.total)
        if self.normalization == "mean":
            denom = self.target_sum / self.total
        elif self.normalization == "range":
            denom = self.target_max - self.target_min
        elif self.normalization == "std":
            denom = torch.sqrt(self.target_squared_sum / self.total - (self.target_sum / self.total) ** 2)
        else:  # l2
            denom = torch.sqrt(self.target_squared_sum)
        return (rmse / torch.abs(denom)).squeeze()

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)
