"""Functional LPIPS / PPL entry points.

Parity: reference functional/image/{lpips,perceptual_path_length}.py. The
reference ships pretrained alex/vgg/squeeze comparators; this environment has
no network, so these functions accept a user-supplied network (``net`` /
``sim_net``) and raise ``ModuleNotFoundError`` without one — the same failure
mode the reference has when torchvision isn't installed.
"""
from __future__ import annotations

from typing import Callable, Optional, Tuple, Union

import torch
from torch import Tensor


def learned_perceptual_image_patch_similarity(
    img1: Tensor,
    img2: Tensor,
    net_type: str = "alex",
    reduction: str = "mean",
    normalize: bool = False,
    net: Optional[Callable] = None,
) -> Tensor:
    """LPIPS distance between two image batches using a user-supplied comparator network."""
    if net is None:
        raise ModuleNotFoundError(
            "learned_perceptual_image_patch_similarity needs a comparator network: pass `net=` a callable"
            " (img1, img2) -> per-sample distances. Pretrained alex/vgg weights cannot be downloaded offline."
        )
    if reduction not in ("mean", "sum"):
        raise ValueError(f"Argument `reduction` must be one of 'mean'/'sum' but got {reduction}")
    loss = net(img1, img2).squeeze()
    return loss.mean() if reduction == "mean" else loss.sum()


def perceptual_path_length(
    generator,
    num_samples: int = 10_000,
    conditional: bool = False,
    batch_size: int = 64,
    interpolation_method: str = "lerp",
    epsilon: float = 1e-4,
    resize: Optional[int] = 64,
    lower_discard: Optional[float] = 0.01,
    upper_discard: Optional[float] = 0.99,
    sim_net: Any = "vgg",
    device: Union[str, torch.device] = "cpu",
) -> Tuple[Tensor, Tensor, Tensor]:
    """Perceptual path length of a generator; returns (mean, std, distances)."""
    from metrics_amd.image.generative import PerceptualPathLength

    metric = PerceptualPathLength(
        num_samples=num_samples,
        conditional=conditional,
        batch_size=batch_size,
        interpolation_method=interpolation_method,
        epsilon=epsilon,
        resize=resize,
        lower_discard=lower_discard,
        upper_discard=upper_discard,
        sim_net=sim_net,
    ).to(device)
    metric.update(generator)
    return metric.compute()
