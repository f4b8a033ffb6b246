"""ctypes bindings for the native COCO matcher (libmetrics_cpu.so)."""
from __future__ import annotations

import ctypes
from typing import List, Optional, Tuple

import numpy as np

from metrics_amd.csrc.build import cpu_lib_path

_LIB: Optional[ctypes.CDLL] = None


def _load() -> Optional[ctypes.CDLL]:
    global _LIB
    if _LIB is not None:
        return _LIB
    p = cpu_lib_path()
    if not p.exists():
        return None
    try:
        _LIB = ctypes.CDLL(str(p))
    except OSError:
        return None
    return _LIB


def native_matcher_available() -> bool:
    return _load() is not None


def _ptr(a: np.ndarray):
    return ctypes.c_void_p(a.ctypes.data)


def coco_eval_class(
    per_img: List[Tuple[np.ndarray, np.ndarray, np.ndarray, np.ndarray, np.ndarray]],
    area_rngs: np.ndarray,
    iou_thrs: np.ndarray,
):
    """Run IoU + greedy matching for one class across images in native code.

    ``per_img``: list of (det_boxes (n,4) f32 score-sorted, det_scores (n,),
    gt_boxes (m,4) f32, gt_crowd (m,), gt_area (m,)).

    Returns ((scores_packed, [(offset, n_det_img)]), dtm[A], dti[A], npig[A]).
    """
    lib = _load()
    n_imgs = len(per_img)
    dt_off = np.zeros(n_imgs + 1, dtype=np.int64)
    gt_off = np.zeros(n_imgs + 1, dtype=np.int64)
    iou_off = np.zeros(n_imgs + 1, dtype=np.int64)
    for i, (db, ds, gb, gc, ga) in enumerate(per_img):
        dt_off[i + 1] = dt_off[i] + db.shape[0]
        gt_off[i + 1] = gt_off[i] + gb.shape[0]
        iou_off[i + 1] = iou_off[i] + db.shape[0] * gb.shape[0]

    total_dt = int(dt_off[-1])
    total_gt = int(gt_off[-1])
    det_boxes = np.ascontiguousarray(
        np.concatenate([p[0] for p in per_img]) if total_dt else np.zeros((0, 4)), dtype=np.float32
    )
    det_scores = np.ascontiguousarray(
        np.concatenate([p[1] for p in per_img]) if total_dt else np.zeros(0), dtype=np.float32
    )
    gt_boxes = np.ascontiguousarray(
        np.concatenate([p[2] for p in per_img]) if total_gt else np.zeros((0, 4)), dtype=np.float32
    )
    gt_crowd = np.ascontiguousarray(
        np.concatenate([p[3] for p in per_img]) if total_gt else np.zeros(0), dtype=np.uint8
    )
    gt_area = np.ascontiguousarray(
        np.concatenate([p[4] for p in per_img]) if total_gt else np.zeros(0), dtype=np.float32
    )
    det_area = (det_boxes[:, 2] - det_boxes[:, 0]) * (det_boxes[:, 3] - det_boxes[:, 1])
    det_area = np.ascontiguousarray(det_area, dtype=np.float32)

    ious = np.zeros(int(iou_off[-1]), dtype=np.float32)
    lib.coco_iou_class(
        _ptr(det_boxes), _ptr(dt_off), _ptr(gt_boxes), _ptr(gt_off), ctypes.c_int64(n_imgs),
        _ptr(gt_crowd), _ptr(ious), _ptr(iou_off),
    )

    A = area_rngs.shape[0]
    T = iou_thrs.shape[0]
    dtm = np.zeros((A, T, total_dt), dtype=np.uint8)
    dti = np.zeros((A, T, total_dt), dtype=np.uint8)
    npig = np.zeros((A, n_imgs), dtype=np.int32)
    area_rngs = np.ascontiguousarray(area_rngs, dtype=np.float32)
    iou_thrs32 = np.ascontiguousarray(iou_thrs, dtype=np.float32)
    lib.coco_match_class(
        _ptr(ious), _ptr(iou_off), _ptr(dt_off), _ptr(gt_off), ctypes.c_int64(n_imgs),
        _ptr(det_area), _ptr(gt_crowd), _ptr(gt_area),
        _ptr(area_rngs), ctypes.c_int64(A),
        _ptr(iou_thrs32), ctypes.c_int64(T),
        _ptr(dtm), _ptr(dti), _ptr(npig),
    )
    offsets = [(int(dt_off[i]), int(dt_off[i + 1] - dt_off[i])) for i in range(n_imgs)]
    return (det_scores, offsets), [dtm[a].astype(bool) for a in range(A)], [dti[a].astype(bool) for a in range(A)], npig.sum(
        axis=1
    )


def coco_eval_class_packed(
    det_boxes: np.ndarray,
    gt_boxes: np.ndarray,
    gt_crowd: np.ndarray,
    gt_area: np.ndarray,
    dt_off: np.ndarray,
    gt_off: np.ndarray,
    area_rngs: np.ndarray,
    iou_thrs: np.ndarray,
):
    """Native IoU + matching on pre-packed per-class arrays.

    Returns (dtm[A] bool (T,total_dt), dti[A], npig[A]).
    """
    lib = _load()
    n_imgs = len(dt_off) - 1
    det_boxes = np.ascontiguousarray(det_boxes, dtype=np.float32)
    gt_boxes = np.ascontiguousarray(gt_boxes, dtype=np.float32)
    gt_crowd = np.ascontiguousarray(gt_crowd, dtype=np.uint8)
    gt_area = np.ascontiguousarray(gt_area, dtype=np.float32)
    dt_off = np.ascontiguousarray(dt_off, dtype=np.int64)
    gt_off = np.ascontiguousarray(gt_off, dtype=np.int64)
    n_dt = np.diff(dt_off)
    n_gt = np.diff(gt_off)
    iou_off = np.concatenate([[0], np.cumsum(n_dt * n_gt)]).astype(np.int64)
    ious = np.zeros(int(iou_off[-1]), dtype=np.float32)
    lib.coco_iou_class(
        _ptr(det_boxes), _ptr(dt_off), _ptr(gt_boxes), _ptr(gt_off), ctypes.c_int64(n_imgs),
        _ptr(gt_crowd), _ptr(ious), _ptr(iou_off),
    )
    det_area = (det_boxes[:, 2] - det_boxes[:, 0]) * (det_boxes[:, 3] - det_boxes[:, 1])
    det_area = np.ascontiguousarray(det_area, dtype=np.float32)
    A = area_rngs.shape[0]
    T = iou_thrs.shape[0]
    total_dt = int(dt_off[-1])
    dtm = np.zeros((A, T, total_dt), dtype=np.uint8)
    dti = np.zeros((A, T, total_dt), dtype=np.uint8)
    npig = np.zeros((A, n_imgs), dtype=np.int32)
    area_rngs = np.ascontiguousarray(area_rngs, dtype=np.float32)
    iou_thrs32 = np.ascontiguousarray(iou_thrs, dtype=np.float32)
    lib.coco_match_class(
        _ptr(ious), _ptr(iou_off), _ptr(dt_off), _ptr(gt_off), ctypes.c_int64(n_imgs),
        _ptr(det_area), _ptr(gt_crowd), _ptr(gt_area),
        _ptr(area_rngs), ctypes.c_int64(A),
        _ptr(iou_thrs32), ctypes.c_int64(T),
        _ptr(dtm), _ptr(dti), _ptr(npig),
    )
    return [dtm[a].astype(bool) for a in range(A)], [dti[a].astype(bool) for a in range(A)], npig.sum(axis=1)
