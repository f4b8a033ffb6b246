"""Segmentation helpers. Parity: torchmetrics ``functional/segmentation/utils.py``.

Morphology (structure elements, erosion), distance transforms and surface /
edge distances used by Hausdorff and the public utils API. The 3D surface-area
neighbour-code table is built from the published deepmind/surface-distance
normal-vector data (see ``_surface_tables.py``).
"""
from __future__ import annotations

import functools
import math
from typing import Optional, Tuple, Union

import torch
from torch import Tensor
from torch.nn.functional import conv2d, conv3d, pad

from metrics_amd.utilities.checks import _check_same_shape


def _ignore_background(preds: Tensor, target: Tensor) -> Tuple[Tensor, Tensor]:
    """Drop the background class (channel 0) from one-hot inputs."""
    preds = preds[:, 1:] if preds.shape[1] > 1 else preds
    target = target[:, 1:] if target.shape[1] > 1 else target
    return preds, target


def check_if_binarized(x: Tensor) -> None:
    """Raise if ``x`` contains anything besides 0/1."""
    if not torch.all(x.bool() == x):
        raise ValueError("Input x should be binarized")


def _unfold(x: Tensor, kernel_size: Tuple[int, ...]) -> Tensor:
    """N-dim sliding windows flattened into dim 1 (for erosion)."""
    for i, k in enumerate(kernel_size):
        x = x.unfold(i + 2, k, 1)
    batch, channels = x.shape[:2]
    spatial = x.shape[2 : 2 + len(kernel_size)]
    return x.reshape(batch, channels, *spatial, -1).flatten(2, 1 + len(kernel_size)).permute(0, 2, 1, 3).flatten(2).permute(0, 2, 1)


def generate_binary_structure(rank: int, connectivity: int) -> Tensor:
    """Structuring element: True where the chebyshev-manhattan distance from the
    center of a 3^rank cube is <= ``connectivity`` (scipy semantics)."""
    if connectivity < 1:
        connectivity = 1
    if rank < 1:
        return torch.tensor([1], dtype=torch.uint8)
    grids = torch.meshgrid([torch.arange(3) for _ in range(rank)], indexing="ij")
    dist = torch.sum(torch.abs(torch.stack(grids, dim=0) - 1), dim=0)
    return dist <= connectivity


def binary_erosion(
    image: Tensor,
    structure: Optional[Tensor] = None,
    origin: Optional[Tuple[int, ...]] = None,
    border_value: int = 0,
) -> Tensor:
    """Binary erosion of a (N, C, H, W) or (N, C, D, H, W) binary image."""
    if not isinstance(image, Tensor):
        raise TypeError(f"Expected argument `image` to be of type Tensor but found {type(image)}")
    if image.ndim not in (4, 5):
        raise ValueError(f"Expected argument `image` to be of rank 4 or 5 but found rank {image.ndim}")
    check_if_binarized(image)

    if structure is None:
        structure = generate_binary_structure(image.ndim - 2, 1).int().to(image.device)
    check_if_binarized(structure)
    if origin is None:
        origin = structure.ndim * (1,)

    if image.is_cuda and image.ndim == 4 and structure.ndim == 2:
        # K8 windowed-min kernel (csrc/ssim.hip): no (N,C,H,W,k^2) unfold
        from metrics_amd.ops import _hip

        return _hip.binary_erosion2d(
            image.to(torch.uint8), structure, (int(origin[0]), int(origin[1])), int(border_value)
        )

    # pad so each output pixel sees its full neighborhood (origin-shifted).
    # F.pad's spec is last-dim-first, so build it reversed — this makes
    # non-square structuring elements work (the reference's un-reversed
    # spec crashes on them; square elements are unaffected)
    pad_spec = [
        x for i in reversed(range(len(origin))) for x in (origin[i], structure.shape[i] - origin[i] - 1)
    ]
    image_pad = pad(image, pad_spec, mode="constant", value=border_value)

    # windows: (N, C, *spatial, prod(kernel)) via unfold
    x = image_pad.float()
    for i, k in enumerate(structure.shape):
        x = x.unfold(i + 2, k, 1)
    window = x.reshape(*x.shape[: 2 + structure.ndim], -1)
    strel = structure.reshape(-1).to(window)
    # eroded pixel = 1 iff every structure-selected neighbor is 1
    result = (window - strel).min(dim=-1).values + 1
    return result.reshape(image.shape).byte()


def distance_transform(
    x: Tensor,
    sampling: Optional[Union[Tensor, list]] = None,
    metric: str = "euclidean",
    engine: str = "pytorch",
) -> Tensor:
    """Distance from each foreground pixel of a 2D binary map to the nearest background pixel."""
    if not isinstance(x, Tensor):
        raise ValueError(f"Expected argument `x` to be of type `torch.Tensor` but got `{type(x)}`.")
    if x.ndim != 2:
        raise ValueError(f"Expected argument `x` to be of rank 2 but got rank `{x.ndim}`.")
    if sampling is not None and not isinstance(sampling, list):
        raise ValueError(
            f"Expected argument `sampling` to either be `None` or of type `list` but got `{type(sampling)}`."
        )
    if metric not in ("euclidean", "chessboard", "taxicab"):
        raise ValueError(
            f"Expected argument `metric` to be one of `['euclidean', 'chessboard', 'taxicab']` but got `{metric}`."
        )
    if engine not in ("pytorch", "scipy"):
        raise ValueError(f"Expected argument `engine` to be one of `['pytorch', 'scipy']` but got `{engine}`.")
    if sampling is None:
        sampling = [1, 1]
    elif len(sampling) != 2:
        raise ValueError(f"Expected argument `sampling` to have length 2 but got length `{len(sampling)}`.")

    if engine == "scipy":
        from scipy import ndimage

        if metric == "euclidean":
            return torch.from_numpy(ndimage.distance_transform_edt(x.cpu().numpy(), sampling))
        return torch.from_numpy(ndimage.distance_transform_cdt(x.cpu().numpy(), sampling, metric=metric).astype(float))

    x = x.float()
    i0, j0 = torch.where(x == 0)
    i1, j1 = torch.where(x == 1)
    if i1.numel() == 0:
        return torch.zeros_like(x)
    dis_row = (i1.view(-1, 1) - i0.view(1, -1)).abs()
    dis_col = (j1.view(-1, 1) - j0.view(1, -1)).abs()
    if metric == "euclidean":
        dis = ((sampling[0] * dis_row) ** 2 + (sampling[1] * dis_col) ** 2).sqrt()
    elif metric == "chessboard":
        dis = torch.max(sampling[0] * dis_row, sampling[1] * dis_col).float()
    else:  # taxicab
        dis = (sampling[0] * dis_row + sampling[1] * dis_col).float()
    h, _ = x.shape
    mindis = dis.min(dim=1).values
    out = torch.zeros_like(x).view(-1)
    out[i1 * h + j1] = mindis
    return out.view(x.shape)


def mask_edges(
    preds: Tensor,
    target: Tensor,
    crop: bool = True,
    spacing: Optional[Union[Tuple[int, int], Tuple[int, int, int]]] = None,
):
    """Edges (and with ``spacing`` also surface areas) of binary masks."""
    _check_same_shape(preds, target)
    if preds.ndim not in (2, 3):
        raise ValueError(f"Expected argument `preds` to be of rank 2 or 3 but got rank `{preds.ndim}`.")
    check_if_binarized(preds)
    check_if_binarized(target)

    if crop:
        or_val = preds | target
        if not or_val.any():
            p, t = torch.zeros_like(preds), torch.zeros_like(target)
            return p, t, p, t
        preds, target = pad(preds, preds.ndim * [1, 1]), pad(target, target.ndim * [1, 1])

    if spacing is None:
        be_pred = binary_erosion(preds.unsqueeze(0).unsqueeze(0)).squeeze() ^ preds
        be_target = binary_erosion(target.unsqueeze(0).unsqueeze(0)).squeeze() ^ target
        return be_pred, be_target

    table, kernel = get_neighbour_tables(spacing, device=preds.device)
    conv_op = conv2d if len(spacing) == 2 else conv3d
    volume = torch.stack([preds.unsqueeze(0), target.unsqueeze(0)], dim=0).float()
    code_preds, code_target = conv_op(volume, kernel.to(volume))
    all_ones = len(table) - 1
    edges_preds = (code_preds != 0) & (code_preds != all_ones)
    edges_target = (code_target != 0) & (code_target != all_ones)
    areas_preds = torch.index_select(table, 0, code_preds.view(-1).int()).view_as(code_preds)
    areas_target = torch.index_select(table, 0, code_target.view(-1).int()).view_as(code_target)
    return edges_preds[0], edges_target[0], areas_preds[0], areas_target[0]


def surface_distance(
    preds: Tensor,
    target: Tensor,
    distance_metric: str = "euclidean",
    spacing: Optional[Union[Tensor, list]] = None,
) -> Tensor:
    """Distance from every edge pixel in ``preds`` to the closest edge in ``target``."""
    if not (preds.dtype == torch.bool and target.dtype == torch.bool):
        raise ValueError(f"Expected both inputs to be of type `torch.bool`, but got {preds.dtype} and {target.dtype}.")
    if not torch.any(target):
        dis = torch.inf * torch.ones_like(target, dtype=torch.float)
    else:
        if not torch.any(preds):
            dis = torch.inf * torch.ones_like(preds, dtype=torch.float)
            return dis[target]
        dis = distance_transform(~target, sampling=spacing, metric=distance_metric)
    return dis[preds]


def edge_surface_distance(
    preds: Tensor,
    target: Tensor,
    distance_metric: str = "euclidean",
    spacing: Optional[Union[Tensor, list]] = None,
    symmetric: bool = False,
):
    """Surface distance between the EDGES of two masks (optionally both directions)."""
    output = mask_edges(preds, target)
    edges_preds, edges_target = output[0].bool(), output[1].bool()
    if symmetric:
        return (
            surface_distance(edges_preds, edges_target, distance_metric=distance_metric, spacing=spacing),
            surface_distance(edges_target, edges_preds, distance_metric=distance_metric, spacing=spacing),
        )
    return surface_distance(edges_preds, edges_target, distance_metric=distance_metric, spacing=spacing)


@functools.lru_cache
def get_neighbour_tables(spacing, device: Optional[torch.device] = None) -> Tuple[Tensor, Tensor]:
    """Neighbour-code -> contour length (2D) or surface area (3D) table + code kernel."""
    if isinstance(spacing, tuple) and len(spacing) == 2:
        return table_contour_length(spacing, device)
    if isinstance(spacing, tuple) and len(spacing) == 3:
        return table_surface_area(spacing, device)
    raise ValueError("The spacing must be a tuple of length 2 or 3.")


def table_contour_length(spacing: Tuple[int, int], device: Optional[torch.device] = None) -> Tuple[Tensor, Tensor]:
    """2D neighbour-code -> contour length table (16 codes, 2x2 kernel)."""
    if not isinstance(spacing, tuple) and len(spacing) != 2:
        raise ValueError("The spacing must be a tuple of length 2.")
    first, second = spacing
    diag = 0.5 * math.sqrt(first**2 + second**2)
    table = torch.zeros(16, dtype=torch.float32, device=device)
    for i in (1, 2, 4, 7, 8, 11, 13, 14):
        table[i] = diag
    for i in (3, 12):
        table[i] = second
    for i in (5, 10):
        table[i] = first
    for i in (6, 9):
        table[i] = 2 * diag
    kernel = torch.as_tensor([[[[8, 4], [2, 1]]]], device=device)
    return table, kernel


@functools.lru_cache
def table_surface_area(spacing: Tuple[int, int, int], device: Optional[torch.device] = None) -> Tuple[Tensor, Tensor]:
    """3D neighbour-code -> surface area table (256 codes, 2x2x2 kernel).

    area(code) = sum over the code's surface normals n of ||n * (s1*s2, s0*s2, s0*s1)||.
    """
    if not isinstance(spacing, tuple) and len(spacing) != 3:
        raise ValueError("The spacing must be a tuple of length 3.")
    from metrics_amd.functional.segmentation._surface_tables import NEIGHBOUR_CODE_TO_NORMALS

    normals = torch.tensor(NEIGHBOUR_CODE_TO_NORMALS, dtype=torch.float32, device=device)  # (256, 4, 3)
    s0, s1, s2 = spacing
    scale = torch.tensor([s1 * s2, s0 * s2, s0 * s1], dtype=torch.float32, device=device)
    table = torch.linalg.vector_norm(normals * scale, dim=-1).sum(-1)
    kernel = torch.as_tensor([[[[[128, 64], [32, 16]], [[8, 4], [2, 1]]]]], device=device)
    return table, kernel
