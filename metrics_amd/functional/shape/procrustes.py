"""Procrustes disparity. Parity: reference functional/shape/procrustes.py."""
from __future__ import annotations

from typing import Tuple, Union

import torch
from torch import Tensor


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
        u, w, v = torch.linalg.svd(
            torch.matmul(point_cloud2.transpose(1, 2), point_cloud1).transpose(1, 2), full_matrices=False
        )
    except Exception as ex:
        raise RuntimeError("SVD calculation in procrustes_disparity did not converge") from ex
    rotation = torch.matmul(u, v)
    scale = w.sum(1, keepdim=True)
    point_cloud2 = scale.unsqueeze(-1) * torch.matmul(point_cloud2, rotation.transpose(1, 2))
    disparity = (point_cloud1 - point_cloud2).square().sum(dim=[1, 2])
    if return_all:
        return disparity, rotation, scale
    return disparity
