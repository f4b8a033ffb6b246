"""Modular Mean squared error. Parity: torchmetrics ``regression/mse.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.regression.mse import _mean_squared_error_compute, _mean_squared_error_update


class MeanSquaredError(Metric):
    """Mean squared error (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    sum_squared_error: Tensor
    total: Tensor

    def __init__(self, squared: bool = True, num_outputs: int = 1, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(squared, bool):
            raise ValueError(f"Expected argument `squared` to be a boolean but got {squared}")
        self.squared = squared
        if not (isinstance(num_outputs, int) and num_outputs > 0):
            raise ValueError(f"Expected num_outputs to be a positive integer but got {num_outputs}")
        self.num_outputs = num_outputs
        self.add_state("sum_squared_error", default=torch.zeros(num_outputs), dist_reduce_fx="sum")
        self.add_state("total", default=torch.tensor(0), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate squared errors (GPU: fused deterministic HIP reduction)."""
        sum_squared_error, num_obs = _mean_squared_error_update(preds, target, self.num_outputs)
        self.sum_squared_error = self.sum_squared_error + sum_squared_error
        self.total = self.total + num_obs

    def compute(self) -> Tensor:
        """MSE (or RMSE if squared=False)."""
        return _mean_squared_error_compute(self.sum_squared_error, self.total, self.squared)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
