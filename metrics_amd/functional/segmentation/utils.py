"""Segmentation helpers. Parity: torchmetrics ``functional/segmentation/utils.py``."""
from __future__ import annotations

from typing import Optional, Tuple, Union

import torch
from torch import Tensor


def _ignore_background(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    """Drop the background class (channel 0) from one-hot inputs."""
    preds = preds[:, 1:] if preds.shape[1] > 1 else preds
    target = target[:, 1:] if target.shape[1] > 1 else target
    return preds, target


def check_if_binarized(x: Tensor) -> None:
    if not torch.all(x.bool() == x):
        raise ValueError("Input x should be binarized")


def _unfold(x: Tensor, kernel_size: Tuple[int, ...]) -> Tensor:
    """N-dim sliding windows (for erosion)."""
    for i, k in enumerate(kernel_size):
        x = x.unfold(i + 2, k, 1)
    return x


def binary_erosion(image: Tensor, kernel_size: int = 3) -> Tensor:
    """Binary erosion via windowed min (conv-style), 2D images (N,1,H,W)."""
    if image.ndim != 4:
        raise ValueError(f"Input image must be 4D (N,C,H,W), got {image.ndim}D")
    check_if_binarized(image)
    pad = kernel_size // 2
    padded = torch.nn.functional.pad(image.float(), [pad] * 4, mode="constant", value=0)
    windows = _unfold(padded, (kernel_size, kernel_size))
    return windows.flatten(-2).min(dim=-1).values


def edge_mask(mask: Tensor) -> Tensor:
    """Boundary pixels of a binary mask (H, W): mask minus its erosion."""
    m = mask[None, None].float()
    eroded = binary_erosion(m)
    return ((m - eroded) > 0)[0, 0]


def surface_distance(
    preds_edge: Tensor, target_edge: Tensor, distance_metric: str = "euclidean",
    spacing: Optional[Union[Tensor, list]] = None,
) -> Tensor:
    """Distances from each pred edge point to the nearest target edge point."""
    if spacing is None:
        spacing = [1, 1]
    sp = torch.as_tensor(spacing, dtype=torch.float, device=preds_edge.device)
    p_pts = torch.nonzero(preds_edge).float() * sp
    t_pts = torch.nonzero(target_edge).float() * sp
    if p_pts.numel() == 0 or t_pts.numel() == 0:
        return torch.tensor([float("inf")], device=preds_edge.device)
    if distance_metric == "euclidean":
        d = torch.cdist(p_pts, t_pts, p=2)
    elif distance_metric == "chessboard":
        d = (p_pts[:, None] - t_pts[None]).abs().max(-1).values
    elif distance_metric == "taxicab":
        d = (p_pts[:, None] - t_pts[None]).abs().sum(-1)
    else:
        raise ValueError(f"Unknown distance_metric {distance_metric}")
    return d.min(dim=1).values
