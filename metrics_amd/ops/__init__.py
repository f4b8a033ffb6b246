"""Kernel dispatch layer.

GPU (HIP/ROCm) tensors route to the in-tree gfx950 kernels (``ops/_hip.py``,
sources in ``csrc/``). CPU tensors use plain torch reference implementations
(these double as the numerics oracles for the GPU kernels in tests).

Policy: a CUDA tensor with the HIP library missing raises — on an MI355X box
the native path must be the one that runs, never a silent eager fallback.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.ops import _hip

hip_available = _hip.hip_available


def hip_bincount(x: Tensor, minlength: int) -> Tensor:
    """GPU histogram (deterministic, LDS-privatized)."""
    return _hip.bincount(x, minlength)


def hip_mc_clf_curve(probs: Tensor, target: Tensor, multilabel: bool = False):
    """Batched per-class exact curves (one composite-key sort) — GPU only."""
    out = _hip.mc_clf_curve(probs, target, multilabel)
    if probs.dtype != torch.float32:
        out = [(f, t, thr.to(probs.dtype), p, n) for f, t, thr, p, n in out]
    return out


def hip_binary_clf_curve(
    preds: Tensor, target: Tensor, weights: Optional[Tensor] = None, pos_label: int = 1
) -> Tuple[Tensor, Tensor, Tensor]:
    """Exact clf-curve core (fps, tps, thresholds) — GPU only (K2 kernel)."""
    return _hip.binary_clf_curve(preds, target, weights, pos_label)


def box_iou_pairwise(boxes1: Tensor, boxes2: Tensor, variant: str = "iou") -> Tensor:
    """All-pairs box IoU with optional GIoU/DIoU/CIoU epilogues (GPU fused, CPU torch)."""
    if boxes1.is_cuda:
        return _hip.box_iou(boxes1, boxes2, variant)
    return _box_iou_torch(boxes1.float(), boxes2.float(), variant)


def _box_iou_torch(b1: Tensor, b2: Tensor, variant: str) -> Tensor:
    area1 = (b1[:, 2] - b1[:, 0]) * (b1[:, 3] - b1[:, 1])
    area2 = (b2[:, 2] - b2[:, 0]) * (b2[:, 3] - b2[:, 1])
    lt = torch.max(b1[:, None, :2], b2[None, :, :2])
    rb = torch.min(b1[:, None, 2:], b2[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    inter = wh[..., 0] * wh[..., 1]
    union = area1[:, None] + area2[None, :] - inter
    iou = torch.where(union > 0, inter / union, torch.zeros_like(union))
    if variant == "iou":
        return iou
    lt_c = torch.min(b1[:, None, :2], b2[None, :, :2])
    rb_c = torch.max(b1[:, None, 2:], b2[None, :, 2:])
    whc = rb_c - lt_c
    if variant == "giou":
        carea = whc[..., 0] * whc[..., 1]
        return torch.where(carea > 0, iou - (carea - union) / carea, iou)
    cdiag = whc[..., 0] ** 2 + whc[..., 1] ** 2 + 1e-7
    cent1 = (b1[:, :2] + b1[:, 2:]) / 2
    cent2 = (b2[:, :2] + b2[:, 2:]) / 2
    dist = ((cent1[:, None, :] - cent2[None, :, :]) ** 2).sum(-1)
    if variant == "diou":
        return iou - dist / cdiag
    # ciou
    w1 = b1[:, 2] - b1[:, 0]
    h1 = b1[:, 3] - b1[:, 1]
    w2 = b2[:, 2] - b2[:, 0]
    h2 = b2[:, 3] - b2[:, 1]
    v = (4 / (torch.pi**2)) * (
        torch.atan(w2[None, :] / (h2[None, :] + 1e-7)) - torch.atan(w1[:, None] / (h1[:, None] + 1e-7))
    ) ** 2
    alpha = v / (1 - iou + v + 1e-7)
    return iou - dist / cdiag - alpha * v


def multiclass_stat_scores_fused(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    ignore_index: Optional[int],
    want_confmat: bool = False,
) -> Tuple[Tensor, Tensor, Tensor, Tensor, Optional[Tensor]]:
    """GPU fused multiclass per-class (tp, fp, tn, fn[, confmat]).

    ``preds`` is (N, C) float logits/probs or (N,) int labels; target (N,) int.
    Returns per-class int64 (C,) tensors. Requires the HIP library (GPU only).
    """
    if preds.ndim == 2 and preds.is_floating_point():
        tp, fp, fn, valid, confmat, _ = _hip.mc_stat_logits(preds, target, ignore_index, want_confmat)
    else:
        tp, fp, fn, valid, confmat = _hip.mc_stat_labels(preds, target, num_classes, ignore_index, want_confmat)
    tn = valid - (tp + fp + fn)
    return tp, fp, tn, fn, confmat


def binary_stat_scores_fused(
    preds: Tensor, target: Tensor, threshold: float, ignore_index: Optional[int]
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """GPU fused binary (tp, fp, tn, fn) over flattened inputs (float preds)."""
    return _hip.binary_stat(preds.flatten(), target.flatten(), threshold, ignore_index)


def multilabel_stat_scores_fused(
    preds: Tensor, target: Tensor, threshold: float, ignore_index: Optional[int]
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """GPU fused per-label (tp, fp, tn, fn) over (N, L) float preds."""
    return _hip.multilabel_stat(preds, target, threshold, ignore_index)


def binary_curve_confmat(
    preds: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int] = None
) -> Tensor:
    """(T,2,2) binary threshold confmat. GPU: bucketized histogram kernel."""
    if preds.is_cuda:
        return _hip.binary_curve_confmat(preds, target, thresholds, ignore_index)
    len_t = len(thresholds)
    preds_t = (preds.unsqueeze(-1) >= thresholds.unsqueeze(0)).long()  # (N, T)
    unique_mapping = preds_t + 2 * target.long().unsqueeze(-1) + 4 * torch.arange(len_t, device=target.device)
    bins = torch.bincount(unique_mapping.flatten(), minlength=4 * len_t)
    return bins.reshape(len_t, 2, 2)


def multiclass_curve_confmat(
    probs: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int] = None
) -> Tensor:
    """(T,C,2,2) one-vs-rest threshold confmats; ``probs`` (B,C) normalized."""
    if probs.is_cuda:
        return _hip.multiclass_curve_confmat(probs, target, thresholds, ignore_index, mode=0)
    len_t = len(thresholds)
    num_classes = probs.shape[1]
    target_t = torch.nn.functional.one_hot(target, num_classes=num_classes)
    preds_t = (probs.unsqueeze(-1) >= thresholds[None, None, :]).long()  # (B, C, T)
    unique_mapping = preds_t + 2 * target_t.unsqueeze(-1)
    unique_mapping += 4 * torch.arange(num_classes, device=probs.device)[None, :, None]
    unique_mapping += 4 * num_classes * torch.arange(len_t, device=probs.device)
    bins = torch.bincount(unique_mapping.flatten(), minlength=4 * num_classes * len_t)
    return bins.reshape(len_t, num_classes, 2, 2)


def err_reduce_sum(x: Tensor, y: Tensor, op: str, eps: float = 1.17e-6) -> Tensor:
    """Fused deterministic Σ f_op(x, y) (fp64). GPU kernel; CPU torch fallback.

    op in {'sq_err','abs_err','ape','sq_log_err','logcosh'} -> scalar;
    'moments' -> (sum_x, sum_x2, sum_y, sum_y2, sum_xy, n).
    """
    if x.is_cuda:
        return _hip.err_reduce(x, y, op, eps)
    xd = x.double().flatten()
    yd = y.double().flatten()
    if op == "sq_err":
        return (xd - yd).pow(2).sum().reshape(1)
    if op == "abs_err":
        return (xd - yd).abs().sum().reshape(1)
    if op == "ape":
        return ((xd - yd).abs() / yd.abs().clamp(min=eps)).sum().reshape(1)
    if op == "sq_log_err":
        return (torch.log1p(xd) - torch.log1p(yd)).pow(2).sum().reshape(1)
    if op == "logcosh":
        d = xd - yd
        return (d + torch.nn.functional.softplus(-2 * d) - torch.log(torch.tensor(2.0))).sum().reshape(1)
    if op == "moments":
        return torch.stack(
            [xd.sum(), (xd * xd).sum(), yd.sum(), (yd * yd).sum(), (xd * yd).sum(), torch.tensor(float(xd.numel()))]
        )
    raise ValueError(f"unknown op {op}")


def multilabel_curve_confmat(
    probs: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int] = None
) -> Tensor:
    """(T,L,2,2) per-label threshold confmats; ``probs`` (B,L) normalized.

    On GPU, elements with target == ignore_index are skipped in-kernel; the
    CPU path expects callers to pre-mask (reference semantics use -1 sentinel
    handled here by clamping counts of negative targets away).
    """
    if probs.is_cuda:
        return _hip.multiclass_curve_confmat(probs, target, thresholds, ignore_index, mode=1)
    len_t = len(thresholds)
    num_labels = probs.shape[1]
    preds_t = (probs.unsqueeze(-1) >= thresholds[None, None, :]).long()
    unique_mapping = preds_t + 2 * target.long().unsqueeze(-1)
    unique_mapping += 4 * torch.arange(num_labels, device=probs.device)[None, :, None]
    unique_mapping += 4 * num_labels * torch.arange(len_t, device=probs.device)
    unique_mapping = unique_mapping[unique_mapping >= 0]
    bins = torch.bincount(unique_mapping.flatten(), minlength=4 * num_labels * len_t)
    return bins.reshape(len_t, num_labels, 2, 2)
