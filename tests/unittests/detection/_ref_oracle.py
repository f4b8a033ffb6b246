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
