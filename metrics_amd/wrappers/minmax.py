"""MinMaxMetric. Parity: torchmetrics ``wrappers/minmax.py``."""
from __future__ import annotations

from typing import Any, Dict

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


class MinMaxMetric(WrapperMetric):
    """Track the min and max of a metric's value across compute calls."""

    full_state_update: bool = True
    min_val: Tensor
    max_val: Tensor

    def __init__(self, base_metric: Metric, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(base_metric, Metric):
            raise ValueError(
                f"Expected base metric to be an instance of `metrics_amd.Metric` but received {base_metric}"
            )
        self._base_metric = base_metric
        # reference parity: min/max are PLAIN attributes, not registered
        # states — they are neither synced nor reset (and forward's
        # full-state dance therefore keeps them running across batches
        # while resetting the base metric each call)
        self.min_val = torch.tensor(float("inf"))
        self.max_val = torch.tensor(float("-inf"))

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Delegate to the base metric."""
        self._base_metric.update(*args, **kwargs)

    def compute(self) -> Dict[str, Tensor]:
        """{'raw', 'min', 'max'} of the base metric."""
        val = self._base_metric.compute()
        if not self._is_suitable_val(val):
            raise RuntimeError(f"Returned value from base metric should be a float or scalar tensor, but got {val}.")
        self.max_val = val if self.max_val.to(val.device) < val else self.max_val.to(val.device)
        self.min_val = val if self.min_val.to(val.device) > val else self.min_val.to(val.device)
        return {"raw": val, "max": self.max_val, "min": self.min_val}

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        # reference parity: run the stock full-state forward dance (which
        # resets the base metric each call — the reference's observable
        # behavior: 'raw' tracks the current batch under forward())
        return Metric.forward(self, *args, **kwargs)

    def reset(self) -> None:
        super().reset()
        self._base_metric.reset()

    @staticmethod
    def _is_suitable_val(val) -> bool:
        if isinstance(val, (int, float)):
            return True
        if isinstance(val, Tensor):
            return val.numel() == 1
        return False

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
