"""MeanAveragePrecision — COCO-style mAP/mAR evaluator, MI355X-native.

Parity: torchmetrics ``detection/mean_ap.py`` public surface (update with
boxes/scores/labels dicts, compute -> map/map_50/map_75/map_small/... dict),
but the evaluator is implemented here from the COCO algorithm itself
(the reference delegates to pycocotools C / faster_coco_eval C++; the
algorithmic spec is torchmetrics ``detection/_mean_ap.py:420-860``):

- all-pairs IoU on device via the HIP ``k_box_iou`` kernel,
- greedy per-image matching at the 10 COCO IoU thresholds with crowd
  handling and area-range ignore flags,
- 101-point interpolated precision accumulation.

Box-format conversion (xyxy/xywh/cxcywh) is implemented directly (the
reference calls torchvision.ops.box_convert).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import torch
from torch import IntTensor, Tensor

from metrics_amd.metric import Metric
from metrics_amd import ops
from metrics_amd.utilities.data import dim_zero_cat


def _fix_empty_tensors(boxes: Tensor) -> Tensor:
    if boxes.numel() == 0 and boxes.ndim == 1:
        return boxes.unsqueeze(0)[:0]
    return boxes


def box_convert(boxes: Tensor, in_fmt: str, out_fmt: str) -> Tensor:
    """Convert between xyxy / xywh / cxcywh box layouts."""
    if in_fmt == out_fmt:
        return boxes.clone()
    if boxes.numel() == 0:
        return boxes
    # to xyxy first
    if in_fmt == "xywh":
        x, y, w, h = boxes.unbind(-1)
        boxes = torch.stack([x, y, x + w, y + h], dim=-1)
    elif in_fmt == "cxcywh":
        cx, cy, w, h = boxes.unbind(-1)
        boxes = torch.stack([cx - w / 2, cy - h / 2, cx + w / 2, cy + h / 2], dim=-1)
    elif in_fmt != "xyxy":
        raise ValueError(f"Unknown box format {in_fmt}")
    if out_fmt == "xyxy":
        return boxes
    x1, y1, x2, y2 = boxes.unbind(-1)
    if out_fmt == "xywh":
        return torch.stack([x1, y1, x2 - x1, y2 - y1], dim=-1)
    if out_fmt == "cxcywh":
        return torch.stack([(x1 + x2) / 2, (y1 + y2) / 2, x2 - x1, y2 - y1], dim=-1)
    raise ValueError(f"Unknown box format {out_fmt}")


def _input_validator(preds, targets, iou_type="bbox") -> None:
    item = "boxes" if iou_type == "bbox" else "masks"
    if not isinstance(preds, Sequence):
        raise ValueError(f"Expected argument `preds` to be of type Sequence, but got {preds}")
    if not isinstance(targets, Sequence):
        raise ValueError(f"Expected argument `target` to be of type Sequence, but got {targets}")
    if len(preds) != len(targets):
        raise ValueError(
            f"Expected argument `preds` and `target` to have the same length, but got {len(preds)} and {len(targets)}"
        )
    for k in (item, "scores", "labels"):
        if any(k not in p for p in preds):
            raise ValueError(f"Expected all dicts in `preds` to contain the `{k}` key")
    for k in (item, "labels"):
        if any(k not in p for p in targets):
            raise ValueError(f"Expected all dicts in `target` to contain the `{k}` key")


class MeanAveragePrecision(Metric):
    """COCO mean average precision / recall for object detection (bbox)."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = True
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    detection_boxes: List[Tensor]
    detection_scores: List[Tensor]
    detection_labels: List[Tensor]
    groundtruth_boxes: List[Tensor]
    groundtruth_labels: List[Tensor]
    groundtruth_crowds: List[Tensor]
    groundtruth_area: List[Tensor]

    def __init__(
        self,
        box_format: str = "xyxy",
        iou_type: str = "bbox",
        iou_thresholds: Optional[List[float]] = None,
        rec_thresholds: Optional[List[float]] = None,
        max_detection_thresholds: Optional[List[int]] = None,
        class_metrics: bool = False,
        extended_summary: bool = False,
        average: str = "macro",
        backend: str = "native",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if box_format not in ("xyxy", "xywh", "cxcywh"):
            raise ValueError(f"Expected argument `box_format` to be one of ('xyxy', 'xywh', 'cxcywh') but got {box_format}")
        if iou_type not in ("bbox",):
            raise ValueError(f"This MI355X-native evaluator supports iou_type='bbox'; got {iou_type} (segm RLE planned)")
        self.box_format = box_format
        self.iou_type = iou_type
        self.iou_thresholds = iou_thresholds or torch.linspace(0.5, 0.95, 10).tolist()
        self.rec_thresholds = rec_thresholds or torch.linspace(0.0, 1.0, 101).tolist()
        self.max_detection_thresholds = sorted(max_detection_thresholds or [1, 10, 100])
        if not isinstance(class_metrics, bool):
            raise ValueError("Expected argument `class_metrics` to be a boolean")
        self.class_metrics = class_metrics
        self.extended_summary = extended_summary
        if average not in ("macro", "micro"):
            raise ValueError(f"Expected argument `average` to be one of ('macro', 'micro') but got {average}")
        self.average = average
        self.backend = backend

        self.add_state("detection_boxes", default=[], dist_reduce_fx=None)
        self.add_state("detection_scores", default=[], dist_reduce_fx=None)
        self.add_state("detection_labels", default=[], dist_reduce_fx=None)
        self.add_state("groundtruth_boxes", default=[], dist_reduce_fx=None)
        self.add_state("groundtruth_labels", default=[], dist_reduce_fx=None)
        self.add_state("groundtruth_crowds", default=[], dist_reduce_fx=None)
        self.add_state("groundtruth_area", default=[], dist_reduce_fx=None)

    # area ranges: all, small, medium, large
    _AREA_RANGES: Tuple[Tuple[float, float], ...] = (
        (0.0, 1e10),
        (0.0, 32**2),
        (32**2, 96**2),
        (96**2, 1e10),
    )

    def update(self, preds: List[Dict[str, Tensor]], target: List[Dict[str, Tensor]]) -> None:
        """Append per-image detections and ground truths."""
        _input_validator(preds, target, self.iou_type)
        for item in preds:
            boxes = _fix_empty_tensors(item["boxes"]).float()
            boxes = box_convert(boxes, in_fmt=self.box_format, out_fmt="xyxy")
            self.detection_boxes.append(boxes)
            self.detection_scores.append(item["scores"].float())
            self.detection_labels.append(item["labels"].long())
        for item in target:
            boxes = _fix_empty_tensors(item["boxes"]).float()
            boxes = box_convert(boxes, in_fmt=self.box_format, out_fmt="xyxy")
            self.groundtruth_boxes.append(boxes)
            self.groundtruth_labels.append(item["labels"].long())
            n = boxes.shape[0]
            crowds = item.get("iscrowd", torch.zeros(n, dtype=torch.long, device=boxes.device)).long()
            self.groundtruth_crowds.append(crowds)
            default_area = (boxes[:, 2] - boxes[:, 0]) * (boxes[:, 3] - boxes[:, 1]) if n else torch.zeros(0, device=boxes.device)
            area = item.get("area", default_area).float()
            if area.numel() == 0 and n:
                area = default_area
            self.groundtruth_area.append(area)

    def _evaluate_image(
        self, det_boxes: Tensor, det_scores: Tensor, gt_boxes: Tensor, gt_crowd: Tensor, gt_area: Tensor,
        area_rng: Tuple[float, float], max_det: int, iou_thrs: Tensor,
    ) -> Optional[Tuple[Tensor, Tensor, Tensor, int]]:
        """COCO evaluateImg for one (image, class): returns (scores, det_matched, det_ignore, n_valid_gt)."""
        n_gt, n_dt = gt_boxes.shape[0], det_boxes.shape[0]
        if n_gt == 0 and n_dt == 0:
            return None
        device = det_boxes.device if n_dt else gt_boxes.device
        T = len(iou_thrs)

        gt_ignore = (gt_crowd == 1) | (gt_area < area_rng[0]) | (gt_area > area_rng[1])
        # sort gts: non-ignored first (stable)
        gt_order = torch.argsort(gt_ignore.to(torch.uint8), stable=True)
        gt_boxes = gt_boxes[gt_order]
        gt_ignore = gt_ignore[gt_order]
        gt_crowd = gt_crowd[gt_order]

        # sort dets by score desc, cap at max_det
        dt_order = torch.argsort(det_scores, descending=True, stable=True)[:max_det]
        det_boxes = det_boxes[dt_order]
        det_scores = det_scores[dt_order]
        n_dt = det_boxes.shape[0]

        if n_dt and n_gt:
            ious = ops.box_iou_pairwise(det_boxes, gt_boxes, "iou")
            # crowd gts use union-over-det IoU semantics (iscrowd => iou vs det area)
            if bool(gt_crowd.any()):
                crowd_cols = torch.nonzero(gt_crowd == 1).flatten()
                if crowd_cols.numel():
                    da = ((det_boxes[:, 2] - det_boxes[:, 0]) * (det_boxes[:, 3] - det_boxes[:, 1])).clamp(min=1e-9)
                    for c in crowd_cols.tolist():
                        g = gt_boxes[c]
                        ix1 = torch.maximum(det_boxes[:, 0], g[0])
                        iy1 = torch.maximum(det_boxes[:, 1], g[1])
                        ix2 = torch.minimum(det_boxes[:, 2], g[2])
                        iy2 = torch.minimum(det_boxes[:, 3], g[3])
                        inter = (ix2 - ix1).clamp(min=0) * (iy2 - iy1).clamp(min=0)
                        ious[:, c] = inter / da
        else:
            ious = torch.zeros(n_dt, n_gt, device=device)

        gt_matched = torch.zeros(T, n_gt, dtype=torch.bool, device=device)
        dt_matched = torch.zeros(T, n_dt, dtype=torch.bool, device=device)
        dt_ignore = torch.zeros(T, n_dt, dtype=torch.bool, device=device)

        # greedy matching (CPU lists — small per (img,cls) sizes)
        ious_c = ious.cpu()
        gt_ignore_c = gt_ignore.cpu().tolist()
        gt_crowd_c = gt_crowd.cpu().tolist()
        for ti, t in enumerate(iou_thrs.tolist()):
            gtm = gt_matched[ti]
            for d in range(n_dt):
                best_iou = min(t, 1 - 1e-10)
                m = -1
                for g in range(n_gt):
                    if gtm[g] and not gt_crowd_c[g]:
                        continue
                    if m > -1 and not gt_ignore_c[m] and gt_ignore_c[g]:
                        break  # gts sorted: once past non-ignored with a match, stop
                    if ious_c[d, g] < best_iou:
                        continue
                    best_iou = ious_c[d, g]
                    m = g
                if m == -1:
                    continue
                dt_ignore[ti, d] = gt_ignore_c[m]
                dt_matched[ti, d] = True
                gt_matched[ti, m] = True

        # unmatched dets outside the area range are ignored
        dt_areas = (det_boxes[:, 2] - det_boxes[:, 0]) * (det_boxes[:, 3] - det_boxes[:, 1])
        dt_out_of_rng = (dt_areas < area_rng[0]) | (dt_areas > area_rng[1])
        dt_ignore = dt_ignore | (~dt_matched & dt_out_of_rng.unsqueeze(0))

        n_valid_gt = int((~gt_ignore).sum())
        return det_scores, dt_matched, dt_ignore, n_valid_gt

    def compute(self) -> Dict[str, Tensor]:
        """COCO mAP/mAR summary over all accumulated images."""
        device = self.detection_boxes[0].device if self.detection_boxes else torch.device("cpu")
        iou_thrs = torch.tensor(self.iou_thresholds, device=device)
        rec_thrs = torch.tensor(self.rec_thresholds, device=device)
        T = len(self.iou_thresholds)
        n_imgs = len(self.detection_boxes)
        max_dets = self.max_detection_thresholds
        max_det_top = max_dets[-1]

        all_labels = (
            torch.cat(self.detection_labels + self.groundtruth_labels)
            if n_imgs
            else torch.zeros(0, dtype=torch.long)
        )
        classes = torch.unique(all_labels).tolist() if all_labels.numel() else []

        A = len(self._AREA_RANGES)
        M = len(max_dets)
        K = len(classes)
        R = len(self.rec_thresholds)
        precision = -torch.ones(T, R, K, A, M, device=device)
        recall = -torch.ones(T, K, A, M, device=device)
        scores_out = -torch.ones(T, R, K, A, M, device=device)

        for ki, cls in enumerate(classes):
            # per image per-class slices
            per_img = []
            for i in range(n_imgs):
                det_m = self.detection_labels[i] == cls
                gt_m = self.groundtruth_labels[i] == cls
                per_img.append((
                    self.detection_boxes[i][det_m],
                    self.detection_scores[i][det_m],
                    self.groundtruth_boxes[i][gt_m],
                    self.groundtruth_crowds[i][gt_m],
                    self.groundtruth_area[i][gt_m],
                ))
            for ai, area_rng in enumerate(self._AREA_RANGES):
                results = [
                    self._evaluate_image(db, ds, gb, gc, ga, area_rng, max_det_top, iou_thrs)
                    for db, ds, gb, gc, ga in per_img
                ]
                results = [r for r in results if r is not None]
                if not results:
                    continue
                for mi, max_det in enumerate(max_dets):
                    scores = torch.cat([r[0][:max_det] for r in results])
                    matched = torch.cat([r[1][:, :max_det] for r in results], dim=1)
                    ignored = torch.cat([r[2][:, :max_det] for r in results], dim=1)
                    npig = sum(r[3] for r in results)
                    if npig == 0:
                        continue
                    order = torch.argsort(scores, descending=True, stable=True)
                    scores_sorted = scores[order]
                    matched = matched[:, order]
                    ignored = ignored[:, order]

                    tps = (matched & ~ignored).float().cumsum(dim=1)
                    fps = (~matched & ~ignored).float().cumsum(dim=1)

                    rc = tps / npig
                    pr = tps / (tps + fps + torch.finfo(torch.float32).eps)

                    if rc.shape[1]:
                        recall[:, ki, ai, mi] = rc[:, -1]
                    else:
                        recall[:, ki, ai, mi] = 0.0

                    # precision envelope (monotone non-increasing from the right)
                    pr_env = pr.flip(1).cummax(dim=1).values.flip(1)
                    # 101-point interpolation: first index where rc >= rec_thr
                    inds = torch.searchsorted(rc.contiguous(), rec_thrs.unsqueeze(0).expand(T, -1).contiguous())
                    for ti in range(T):
                        row = pr_env[ti]
                        srow = scores_sorted
                        idx = inds[ti]
                        valid = idx < row.shape[0]
                        q = torch.zeros(R, device=device)
                        s = torch.zeros(R, device=device)
                        q[valid] = row[idx[valid]]
                        s[valid] = srow[idx[valid]]
                        precision[ti, :, ki, ai, mi] = q
                        scores_out[ti, :, ki, ai, mi] = s

        def _summarize(ap: bool, iou_thr: Optional[float] = None, area: int = 0, max_det_idx: int = -1) -> Tensor:
            if ap:
                s = precision[:, :, :, area, max_det_idx]
                if iou_thr is not None:
                    ti = self.iou_thresholds.index(iou_thr)
                    s = s[ti : ti + 1]
            else:
                s = recall[:, :, area, max_det_idx]
                if iou_thr is not None:
                    ti = self.iou_thresholds.index(iou_thr)
                    s = s[ti : ti + 1]
            valid = s > -1
            if valid.sum() == 0:
                return torch.tensor(-1.0, device=device)
            return s[valid].mean()

        result: Dict[str, Tensor] = {}
        result["map"] = _summarize(True)
        result["map_50"] = _summarize(True, 0.5) if 0.5 in self.iou_thresholds else torch.tensor(-1.0)
        result["map_75"] = _summarize(True, 0.75) if 0.75 in self.iou_thresholds else torch.tensor(-1.0)
        result["map_small"] = _summarize(True, area=1)
        result["map_medium"] = _summarize(True, area=2)
        result["map_large"] = _summarize(True, area=3)
        for mi, md in enumerate(max_dets):
            result[f"mar_{md}"] = _summarize(False, max_det_idx=mi)
        result["mar_small"] = _summarize(False, area=1)
        result["mar_medium"] = _summarize(False, area=2)
        result["mar_large"] = _summarize(False, area=3)

        if self.class_metrics and classes:
            map_per_class = []
            mar_per_class = []
            for ki in range(K):
                p = precision[:, :, ki, 0, -1]
                v = p[p > -1]
                map_per_class.append(v.mean() if v.numel() else torch.tensor(-1.0, device=device))
                r = recall[:, ki, 0, -1]
                v = r[r > -1]
                mar_per_class.append(v.mean() if v.numel() else torch.tensor(-1.0, device=device))
            result["map_per_class"] = torch.stack(map_per_class)
            result[f"mar_{max_dets[-1]}_per_class"] = torch.stack(mar_per_class)
        else:
            result["map_per_class"] = torch.tensor(-1.0, device=device)
            result[f"mar_{max_dets[-1]}_per_class"] = torch.tensor(-1.0, device=device)
        result["classes"] = torch.tensor(classes, dtype=torch.int, device=device)

        if self.extended_summary:
            result["precision"] = precision
            result["recall"] = recall
            result["scores"] = scores_out
            result["ious"] = torch.tensor([], device=device)  # per-pair ious not retained
        return result

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)

    def _sync_dist(self, dist_sync_fn=None, process_group=None) -> None:
        """List-of-variable-shape states: gather each element (reference uses all_gather_object)."""
        super()._sync_dist(dist_sync_fn=dist_sync_fn, process_group=process_group)
