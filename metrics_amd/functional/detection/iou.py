"""Box IoU family (functional).

Parity: torchmetrics ``functional/detection/{iou,giou,diou,ciou}.py``.
Backed by the in-tree HIP all-pairs kernel (csrc/kernels.hip k_box_iou) with
GIoU/DIoU/CIoU epilogues; the reference delegates to torchvision C++/CUDA ops.
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from metrics_amd import ops


def _iou_variant_fn(variant: str):
    def _update(preds: Tensor, target: Tensor, iou_threshold: Optional[float], replacement_val: float = 0) -> Tensor:
        iou = ops.box_iou_pairwise(preds, target, variant)
        if iou_threshold is not None:
            iou[iou < iou_threshold] = replacement_val
        return iou

    def _compute(iou: Tensor, aggregate: bool = True) -> Tensor:
        if not aggregate:
            return iou
        return iou.diag().mean() if iou.numel() > 0 else torch.tensor(0.0, device=iou.device)

    def fn(
        preds: Tensor,
        target: Tensor,
        iou_threshold: Optional[float] = None,
        replacement_val: float = 0,
        aggregate: bool = True,
    ) -> Tensor:
        if not isinstance(preds, Tensor) or not isinstance(target, Tensor):
            raise ValueError("Expected both `preds` and `target` to be tensors of shape (N, 4) xyxy boxes")
        iou = _update(preds, target, iou_threshold, replacement_val)
        return _compute(iou, aggregate)

    return _update, _compute, fn


_iou_update, _iou_compute, intersection_over_union = _iou_variant_fn("iou")
_giou_update, _giou_compute, generalized_intersection_over_union = _iou_variant_fn("giou")
_diou_update, _diou_compute, distance_intersection_over_union = _iou_variant_fn("diou")
_ciou_update, _ciou_compute, complete_intersection_over_union = _iou_variant_fn("ciou")
intersection_over_union.__name__ = "intersection_over_union"
generalized_intersection_over_union.__name__ = "generalized_intersection_over_union"
distance_intersection_over_union.__name__ = "distance_intersection_over_union"
complete_intersection_over_union.__name__ = "complete_intersection_over_union"
