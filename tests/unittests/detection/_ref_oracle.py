"""Load the reference's pure-torch COCOeval (`detection/_mean_ap.py`) as an
offline oracle for differential mAP fuzzing.

The legacy reference class only *algorithmically* needs box math for
iou_type="bbox"; its imports of torchvision/pycocotools are satisfied with
local stubs (our own implementations of box_area/box_iou/box_convert; mask
utils raise — segm is never exercised through this oracle). Returns None
when /root/reference is not present (e.g. on a GPU box) so callers skip.
"""
from __future__ import annotations

import importlib.machinery
import os
import sys
import types

import torch

_REF_SRC = "/root/reference/src"
_SHIM = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", "..", "tools", "refbench")


def _box_area(b):
    # no clamping — matches torchvision.ops.box_area (degenerate boxes get
    # negative area, which the area-range ignore logic relies on)
    return (b[:, 2] - b[:, 0]) * (b[:, 3] - b[:, 1])


def _box_iou(a, b):
    area1, area2 = _box_area(a), _box_area(b)
    lt = torch.max(a[:, None, :2], b[None, :, :2])
    rb = torch.min(a[:, None, 2:], b[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    inter = wh[..., 0] * wh[..., 1]
    union = area1[:, None] + area2[None, :] - inter
    return torch.where(union > 0, inter / union, torch.zeros_like(union))


def _box_convert(boxes, in_fmt, out_fmt):
    assert out_fmt == "xyxy"
    if in_fmt == "xyxy":
        return boxes
    if in_fmt == "xywh":
        out = boxes.clone()
        out[:, 2:] = boxes[:, :2] + boxes[:, 2:]
        return out
    if in_fmt == "cxcywh":
        out = boxes.clone()
        out[:, :2] = boxes[:, :2] - boxes[:, 2:] / 2
        out[:, 2:] = boxes[:, :2] + boxes[:, 2:] / 2
        return out
    raise ValueError(in_fmt)


def _box_iou_variant(a, b, variant):
    iou = _box_iou(a, b)
    lt_c = torch.min(a[:, None, :2], b[None, :, :2])
    rb_c = torch.max(a[:, None, 2:], b[None, :, 2:])
    whc = rb_c - lt_c
    area1, area2 = _box_area(a), _box_area(b)
    lt = torch.max(a[:, None, :2], b[None, :, :2])
    rb = torch.min(a[:, None, 2:], b[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    union = area1[:, None] + area2[None, :] - wh[..., 0] * wh[..., 1]
    if variant == "giou":
        carea = whc[..., 0] * whc[..., 1]
        return torch.where(carea > 0, iou - (carea - union) / carea, iou)
    cdiag = whc[..., 0] ** 2 + whc[..., 1] ** 2 + 1e-7
    c1 = (a[:, :2] + a[:, 2:]) / 2
    c2 = (b[:, :2] + b[:, 2:]) / 2
    dist = ((c1[:, None, :] - c2[None, :, :]) ** 2).sum(-1)
    if variant == "diou":
        return iou - dist / cdiag
    w1, h1 = a[:, 2] - a[:, 0], a[:, 3] - a[:, 1]
    w2, h2 = b[:, 2] - b[:, 0], b[:, 3] - b[:, 1]
    import math

    v = (4 / math.pi**2) * (
        torch.atan(w2[None, :] / (h2[None, :] + 1e-7)) - torch.atan(w1[:, None] / (h1[:, None] + 1e-7))
    ) ** 2
    alpha = v / (1 - iou + v + 1e-7)
    return iou - dist / cdiag - alpha * v


def _stub(name: str) -> types.ModuleType:
    mod = types.ModuleType(name)
    mod.__spec__ = importlib.machinery.ModuleSpec(name, None)
    sys.modules[name] = mod
    return mod


def load_legacy_map():
    """Return the reference's pure-torch MeanAveragePrecision class, or None."""
    if not os.path.isdir(_REF_SRC):
        return None
    if _REF_SRC not in sys.path:
        sys.path.insert(0, os.path.abspath(_SHIM))
        sys.path.insert(0, _REF_SRC)
    if "torchvision" not in sys.modules:
        tv = _stub("torchvision")
        tv.__version__ = "0.20.0"
        ops = _stub("torchvision.ops")
        ops.box_area = _box_area
        ops.box_iou = _box_iou
        ops.box_convert = _box_convert
        # giou/diou/ciou variants (standard formulas; used by the module-level
        # IoU differential, which tests AGGREGATION semantics — the box math
        # itself is covered by hand-computed cases in the detection tests)
        ops.generalized_box_iou = lambda a, b: _box_iou_variant(a, b, "giou")
        ops.distance_box_iou = lambda a, b: _box_iou_variant(a, b, "diou")
        ops.complete_box_iou = lambda a, b: _box_iou_variant(a, b, "ciou")
        tv.ops = ops
    if "pycocotools" not in sys.modules:
        pc = _stub("pycocotools")
        mask = _stub("pycocotools.mask")

        def _no_masks(*_a, **_k):
            raise RuntimeError("segm path not supported by the offline oracle")

        mask.encode = mask.decode = mask.area = mask.iou = _no_masks
        pc.mask = mask
    try:
        import torchmetrics.detection._mean_ap as legacy
    except Exception:
        return None
    legacy._PYCOCOTOOLS_AVAILABLE = True
    legacy._TORCHVISION_AVAILABLE = True
    # The legacy class deviates from pycocotools in the greedy matching rule:
    # it forbids matching area-ignored gts, uses a strict `>` threshold and
    # argmax (first-max) tie-breaking. Swap in a pycocotools-faithful rule
    # (written here, from the published COCOeval algorithm) so the oracle's
    # *independent* accumulation machinery checks our engine without those
    # known drifts: iterate gts sorted ignore-last, stop at the first ignored
    # gt once a non-ignored match exists, accept IoU >= threshold, later
    # equal-IoU gts win.
    def _coco_find_best_gt_match(threshold, gt_matches, idx_iou, gt_ignore, ious, idx_det):
        best = min(threshold, 1 - 1e-10)
        m = -1
        for g in range(gt_ignore.numel()):
            if bool(gt_matches[idx_iou, g]):
                continue
            if m > -1 and not bool(gt_ignore[m]) and bool(gt_ignore[g]):
                break
            v = float(ious[idx_det, g])
            if v < best:
                continue
            best = v
            m = g
        return m

    legacy.MeanAveragePrecision._find_best_gt_match = staticmethod(_coco_find_best_gt_match)
    return legacy.MeanAveragePrecision
