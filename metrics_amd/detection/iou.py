"""Modular IoU metrics over detection dicts.

Parity: torchmetrics ``detection/{iou,giou,diou,ciou}.py`` — update takes
lists of ``{"boxes", "labels"[, "scores"]}`` dicts; when ``respect_labels``
only same-label pairs count.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd import ops


def _input_validator_iou(preds, target, ignore_score: bool = False) -> None:
    if not isinstance(preds, (list, tuple)) or not isinstance(target, (list, tuple)):
        raise ValueError("Expected `preds` and `target` to be a sequence of dicts")
    if len(preds) != len(target):
        raise ValueError("Expected argument `preds` and `target` to have the same length")
    for k in ("boxes",) + (() if ignore_score else ("scores",)):
        if any(k not in p for p in preds):
            raise ValueError(f"Expected all dicts in `preds` to contain the `{k}` key")
    if any("boxes" not in t for t in target):
        raise ValueError("Expected all dicts in `target` to contain the `boxes` key")
    if any("labels" not in p for p in preds):
        raise ValueError("Expected all dicts in `preds` to contain the `labels` key")
    if any("labels" not in t for t in target):
        raise ValueError("Expected all dicts in `target` to contain the `labels` key")


class IntersectionOverUnion(Metric):
    """Mean IoU between matched predicted and ground-truth boxes."""

    is_differentiable = False
    higher_is_better = True
    full_state_update: bool = True
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    _iou_variant: str = "iou"
    _invalid_val: float = -1.0

    def __init__(
        self,
        box_format: str = "xyxy",
        iou_threshold: Optional[float] = None,
        class_metrics: bool = False,
        respect_labels: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if box_format not in ("xyxy", "xywh", "cxcywh"):
            raise ValueError(f"Expected argument `box_format` to be one of ('xyxy', 'xywh', 'cxcywh') but got {box_format}")
        self.box_format = box_format
        if iou_threshold is not None and not isinstance(iou_threshold, float):
            raise ValueError(f"Expected argument `iou_threshold` to be a float or None, but got {iou_threshold}")
        self.iou_threshold = iou_threshold
        if not isinstance(class_metrics, bool):
            raise ValueError("Expected argument `class_metrics` to be a boolean")
        self.class_metrics = class_metrics
        if not isinstance(respect_labels, bool):
            raise ValueError("Expected argument `respect_labels` to be a boolean")
        self.respect_labels = respect_labels

        self.add_state("groundtruth_labels", default=[], dist_reduce_fx=None)
        self.add_state("iou_matrix", default=[], dist_reduce_fx=None)

    def update(self, preds: List[Dict[str, Tensor]], target: List[Dict[str, Tensor]]) -> None:
        """Accumulate the label-masked all-pairs IoU matrices (reference
        semantics: compute() means over every pair whose entry survived the
        label/threshold masking — NOT a best-match assignment)."""
        _input_validator_iou(preds, target, ignore_score=True)
        from metrics_amd.detection.mean_ap import box_convert

        for p, t in zip(preds, target):
            p_boxes, p_labels = p["boxes"].float(), p["labels"]
            t_boxes, t_labels = t["boxes"].float(), t["labels"]
            if self.box_format != "xyxy":
                if p_boxes.numel():
                    p_boxes = box_convert(p_boxes, in_fmt=self.box_format, out_fmt="xyxy")
                if t_boxes.numel():
                    t_boxes = box_convert(t_boxes, in_fmt=self.box_format, out_fmt="xyxy")
            self.groundtruth_labels.append(t_labels)
            # reference empty-input quirks (functional/detection/iou.py:35-38):
            # no preds -> zeros(M, M); no gts -> zeros(N, N)
            if p_boxes.numel() == 0:
                iou = torch.zeros(t_boxes.shape[0], t_boxes.shape[0], device=t_boxes.device)
            elif t_boxes.numel() == 0:
                iou = torch.zeros(p_boxes.shape[0], p_boxes.shape[0], device=p_boxes.device)
            else:
                iou = ops.box_iou_pairwise(p_boxes, t_boxes, self._iou_variant)
            if self.iou_threshold is not None:
                iou[iou < self.iou_threshold] = self._invalid_val
            if self.respect_labels:
                if p_boxes.numel() > 0 and t_boxes.numel() > 0:
                    label_eq = p_labels.unsqueeze(1) == t_labels.unsqueeze(0)
                else:
                    label_eq = torch.eye(iou.shape[0], dtype=torch.bool, device=iou.device)
                iou[~label_eq] = self._invalid_val
            self.iou_matrix.append(iou)

    def compute(self) -> Dict[str, Tensor]:
        """Mean (per-class) IoU over all surviving pairs."""
        name = self._iou_variant
        if not self.iou_matrix:
            return {name: torch.tensor(0.0)}
        valid = [m[m != self._invalid_val] for m in self.iou_matrix]
        score = torch.cat(valid, 0).mean() if valid else torch.tensor(0.0)
        result = {name: score}
        if torch.isnan(score):
            result[name] = torch.tensor(0.0, device=score.device)
        if self.class_metrics:
            from metrics_amd.utilities.data import dim_zero_cat

            gt_labels = dim_zero_cat(self.groundtruth_labels)
            classes = gt_labels.unique().tolist() if len(gt_labels) > 0 else []
            for cl in classes:
                masked = torch.zeros_like(score)
                observed = torch.zeros_like(score)
                for mat, gt_lab in zip(self.iou_matrix, self.groundtruth_labels):
                    if mat.shape[-1] != gt_lab.shape[0]:
                        continue  # empty-input square quirk: columns are not gts
                    scores_cl = mat[:, gt_lab == cl]
                    masked += scores_cl[scores_cl != self._invalid_val].sum()
                    observed += (scores_cl != self._invalid_val).sum()
                result[f"{name}/cl_{cl}"] = masked / observed
        return result

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class GeneralizedIntersectionOverUnion(IntersectionOverUnion):
    """Mean GIoU between matched boxes."""

    _iou_variant = "giou"
    _invalid_val = -1.0
    plot_lower_bound: float = -1.0


class DistanceIntersectionOverUnion(IntersectionOverUnion):
    """Mean DIoU between matched boxes."""

    _iou_variant = "diou"
    _invalid_val = -1.0
    plot_lower_bound: float = -1.0


class CompleteIntersectionOverUnion(IntersectionOverUnion):
    """Mean CIoU between matched boxes."""

    _iou_variant = "ciou"
    _invalid_val = -2.0
    plot_lower_bound: float = -1.0
