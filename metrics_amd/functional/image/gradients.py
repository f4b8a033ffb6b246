"""Finite-difference image gradients. Parity: reference functional/image/gradients.py:45."""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor


def image_gradients(img: Tensor) -> Tuple[Tensor, Tensor]:
    """Return (dy, dx) one-step forward differences, zero-padded at the far edge."""
    if not isinstance(img, Tensor):
        raise TypeError(f"The `img` expects a value of <Tensor> type but got {type(img)}")
    if img.ndim != 4:
        raise RuntimeError(f"The `img` expects a 4D tensor but got {img.ndim}D tensor")
    dy = img[..., 1:, :] - img[..., :-1, :]
    dx = img[..., :, 1:] - img[..., :, :-1]
    shapey = [img.shape[0], img.shape[1], 1, img.shape[3]]
    dy = torch.cat([dy, torch.zeros(shapey, device=img.device, dtype=img.dtype)], dim=2)
    dy = dy.view(img.shape)
    shapex = [img.shape[0], img.shape[1], img.shape[2], 1]
    dx = torch.cat([dx, torch.zeros(shapex, device=img.device, dtype=img.dtype)], dim=3)
    dx = dx.view(img.shape)
    return dy, dx
