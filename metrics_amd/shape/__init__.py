"""Shape metrics. Parity: torchmetrics ``shape/procrustes.py``."""
from __future__ import annotations

from typing import Any, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.shape import procrustes_disparity


class ProcrustesDisparity(Metric):
    """Procrustes disparity (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    disparity: Tensor
    total: Tensor

    def __init__(self, reduction: str = "mean", **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if reduction not in ("mean", "sum"):
            raise ValueError(f"Argument `reduction` must be one of ['mean', 'sum'], got {reduction}")
        self.reduction = reduction
        self.add_state("disparity", default=torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("total", default=torch.tensor(0), dist_reduce_fx="sum")

    def update(self, point_cloud1: Tensor, point_cloud2: Tensor) -> None:
        """Accumulate per-cloud disparities."""
        disparity = procrustes_disparity(point_cloud1, point_cloud2)
        self.disparity += disparity.sum()
        self.total += disparity.numel()

    def compute(self) -> Tensor:
        if self.reduction == "mean":
            return self.disparity / self.total
        return self.disparity

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


__all__ = ["ProcrustesDisparity", "procrustes_disparity"]
