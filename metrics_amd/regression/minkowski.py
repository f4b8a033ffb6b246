"""Modular Minkowski distance. Parity: torchmetrics ``regression/minkowski.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.regression.minkowski import _minkowski_distance_compute, _minkowski_distance_update


class MinkowskiDistance(Metric):
    """Minkowski distance (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    minkowski_dist_sum: Tensor

    def __init__(self, p: float, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        from metrics_amd.utilities.exceptions import MetricsUserError

        if not (isinstance(p, (float, int)) and p >= 1):
            raise MetricsUserError(f"Argument ``p`` must be a float or int greater than 1, but got {p}")
        self.p = p
        self.add_state("minkowski_dist_sum", default=torch.tensor(0.0), dist_reduce_fx="sum")

    def update(self, preds: Tensor, targets: Tensor) -> None:
        """Accumulate |p - t|^p sums."""
        self.minkowski_dist_sum = self.minkowski_dist_sum + _minkowski_distance_update(preds, targets, self.p)

    def compute(self) -> Tensor:
        return _minkowski_distance_compute(self.minkowski_dist_sum, self.p)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
