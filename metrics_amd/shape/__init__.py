"""Shape metrics. Parity: torchmetrics ``shape/procrustes.py``."""
from __future__ import annotations

from typing import Any, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric


def procrustes_disparity(
    point_cloud1: Tensor, point_cloud2: Tensor, return_all: bool = False
) -> Union[Tensor, Tuple[Tensor, Tensor, Tensor]]:
    """Procrustes disparity: residual after optimal translation/scale/rotation alignment (SVD)."""
    if point_cloud1.shape != point_cloud2.shape:
        raise ValueError("Expected both datasets to have the same shape")
    if point_cloud1.ndim != 3:
        raise ValueError("Expected both datasets to be 3D tensors of shape (N, M, D)")

    point_cloud1 = point_cloud1 - point_cloud1.mean(dim=1, keepdim=True)
    point_cloud2 = point_cloud2 - point_cloud2.mean(dim=1, keepdim=True)
    point_cloud1 = point_cloud1 / torch.linalg.norm(point_cloud1, dim=[1, 2], keepdim=True)
    point_cloud2 = point_cloud2 / torch.linalg.norm(point_cloud2, dim=[1, 2], keepdim=True)

    try:
        u, w, v = torch.linalg.svd(torch.matmul(point_cloud2.transpose(1, 2), point_cloud1).transpose(1, 2), full_matrices=False)
    except Exception as ex:
        raise RuntimeError("SVD calculation in procrustes_disparity did not converge") from ex
    rotation = torch.matmul(u, v)
    scale = w.sum(1, keepdim=True)
    point_cloud2 = scale.unsqueeze(-1) * torch.matmul(point_cloud2, rotation.transpose(1, 2))
    disparity = (point_cloud1 - point_cloud2).square().sum(dim=[1, 2])
    if return_all:
        return disparity, rotation, scale
    return disparity


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
