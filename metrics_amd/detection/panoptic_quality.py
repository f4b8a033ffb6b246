"""Panoptic quality (+ modified PQ).

Parity: torchmetrics ``detection/panoptic_qualities.py`` /
``functional/detection/_panoptic_quality_common.py``: segment matching at
IoU > 0.5 per category with void handling; PQ = sum_iou / (TP + FP/2 + FN/2).
"""
from __future__ import annotations

from typing import Any, Collection, Dict, List, Optional, Set, Tuple

import torch
from torch import Tensor

from metrics_amd.metric import Metric


def _parse_categories(things: Collection[int], stuffs: Collection[int]) -> Tuple[Set[int], Set[int]]:
    things = set(int(t) for t in things)
    stuffs = set(int(s) for s in stuffs)
    if things & stuffs:
        raise ValueError(f"Expected arguments `things` and `stuffs` to have distinct keys, but got {things} and {stuffs}")
    if not (things | stuffs):
        raise ValueError("At least one of `things` and `stuffs` must be non-empty.")
    return things, stuffs


def _validate_inputs(preds: Tensor, target: Tensor) -> None:
    if not isinstance(preds, Tensor) or not isinstance(target, Tensor):
        raise TypeError("Expected argument `preds` and `target` to be of type `torch.Tensor`")
    if preds.shape != target.shape:
        raise ValueError(
            f"Expected argument `preds` and `target` to have the same shape, but got {preds.shape} and {target.shape}"
        )
    if preds.dim() < 3 or preds.shape[-1] != 2:
        raise ValueError(
            "Expected argument `preds` to have at least 3 dimensions and the final dimension equal to 2"
        )


def _flat_colors(pan: Tensor, things: Set[int], stuffs: Set[int], void_cat: int) -> Tuple[Tensor, Tensor]:
    """Flatten to per-pixel (category, instance); stuff instances collapse to 0,
    unknown categories map to the void color ``(void_cat, 0)``."""
    cats = pan[..., 0].flatten().clone()
    insts = pan[..., 1].flatten().clone()
    dev = pan.device
    is_stuff = torch.isin(cats, torch.tensor(sorted(stuffs), device=dev))
    insts[is_stuff] = 0
    unknown = ~torch.isin(cats, torch.tensor(sorted(things | stuffs), device=dev))
    cats[unknown] = void_cat
    insts[unknown] = 0
    return cats, insts


def _panoptic_quality_update_sample(
    pred: Tensor,
    target: Tensor,
    things: Set[int],
    stuffs: Set[int],
    modified_metric_stuffs: Optional[Set[int]] = None,
) -> Tuple[Dict[int, float], Dict[int, int], Dict[int, int], Dict[int, int]]:
    """One image: per-category (iou_sum, tp, fp, fn).

    Matching follows the published PQ algorithm exactly as the reference
    implements it (functional/detection/_panoptic_quality_common.py:312):
    every same-category (pred segment, target segment) intersection with
    IoU > 0.5 is a TP (the gate makes matches exclusive); the union subtracts
    the pred segment's overlap with void target AND the target segment's
    overlap with void pred; unmatched segments that are majority-void in the
    other map are ignored rather than counted as FP/FN. For the modified PQ
    variant, stuff classes accumulate IoU at any overlap > 0, count one TP
    per target segment present, and contribute no FP/FN.
    """
    mod = modified_metric_stuffs or set()
    void_cat = max(things | stuffs) + 1
    VOID = (void_cat, 0)
    pc, pi = _flat_colors(pred, things, stuffs, void_cat)
    tc, ti = _flat_colors(target, things, stuffs, void_cat)

    # one unique pass over per-pixel (pred color, target color) rows gives the
    # pairwise intersection areas; segment areas are its row/col marginals
    rows = torch.stack([pc, pi, tc, ti], dim=1)
    uniq, counts = torch.unique(rows, dim=0, return_counts=True)
    pair_area: Dict[Tuple[Tuple[int, int], Tuple[int, int]], int] = {}
    pred_area: Dict[Tuple[int, int], int] = {}
    target_area: Dict[Tuple[int, int], int] = {}
    for (a, b, c, d), n in zip(uniq.tolist(), counts.tolist()):
        pcol, tcol = (a, b), (c, d)
        pair_area[(pcol, tcol)] = pair_area.get((pcol, tcol), 0) + n
        pred_area[pcol] = pred_area.get(pcol, 0) + n
        target_area[tcol] = target_area.get(tcol, 0) + n

    iou_sum: Dict[int, float] = {}
    tp: Dict[int, int] = {}
    fp: Dict[int, int] = {}
    fn: Dict[int, int] = {}
    matched_pred: Set[Tuple[int, int]] = set()
    matched_target: Set[Tuple[int, int]] = set()

    for (pcol, tcol), inter in pair_area.items():
        if tcol == VOID or pcol == VOID or pcol[0] != tcol[0]:
            continue
        cat = tcol[0]
        union = (
            pred_area[pcol]
            - pair_area.get((pcol, VOID), 0)
            + target_area[tcol]
            - pair_area.get((VOID, tcol), 0)
            - inter
        )
        iou = inter / union if union > 0 else 0.0
        if cat not in mod and iou > 0.5:
            matched_pred.add(pcol)
            matched_target.add(tcol)
            iou_sum[cat] = iou_sum.get(cat, 0.0) + iou
            tp[cat] = tp.get(cat, 0) + 1
        elif cat in mod and iou > 0:
            iou_sum[cat] = iou_sum.get(cat, 0.0) + iou

    for tcol in set(target_area) - matched_target:
        if tcol == VOID or tcol[0] in mod:
            continue
        if pair_area.get((VOID, tcol), 0) / target_area[tcol] <= 0.5:
            fn[tcol[0]] = fn.get(tcol[0], 0) + 1

    for pcol in set(pred_area) - matched_pred:
        if pcol == VOID or pcol[0] in mod:
            continue
        if pair_area.get((pcol, VOID), 0) / pred_area[pcol] <= 0.5:
            fp[pcol[0]] = fp.get(pcol[0], 0) + 1

    for tcol in target_area:
        if tcol != VOID and tcol[0] in mod:
            tp[tcol[0]] = tp.get(tcol[0], 0) + 1

    return iou_sum, tp, fp, fn


class PanopticQuality(Metric):
    """Panoptic quality for (category, instance)-encoded panoptic maps."""

    is_differentiable = False
    higher_is_better = True
    full_state_update = False
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    iou_sum: Tensor
    true_positives: Tensor
    false_positives: Tensor
    false_negatives: Tensor

    _modified: bool = False

    def __init__(
        self,
        things: Collection[int],
        stuffs: Collection[int],
        allow_unknown_preds_category: bool = False,
        return_sq_and_rq: bool = False,
        return_per_class: bool = False,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.things, self.stuffs = _parse_categories(things, stuffs)
        self.allow_unknown_preds_category = allow_unknown_preds_category
        self.return_sq_and_rq = return_sq_and_rq
        self.return_per_class = return_per_class
        cats = sorted(self.things | self.stuffs)
        self._cat_to_idx = {c: i for i, c in enumerate(cats)}
        self._cats = cats
        n = len(cats)
        self.add_state("iou_sum", default=torch.zeros(n, dtype=torch.double), dist_reduce_fx="sum")
        self.add_state("true_positives", default=torch.zeros(n, dtype=torch.long), dist_reduce_fx="sum")
        self.add_state("false_positives", default=torch.zeros(n, dtype=torch.long), dist_reduce_fx="sum")
        self.add_state("false_negatives", default=torch.zeros(n, dtype=torch.long), dist_reduce_fx="sum")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Match segments per image and accumulate per-category counts."""
        _validate_inputs(preds, target)
        if not self.allow_unknown_preds_category:
            known = torch.tensor(self._cats, device=preds.device)
            unknown = ~torch.isin(preds[..., 0], known)
            if bool(unknown.any()):
                raise ValueError(
                    "Unknown categories found in `preds`. Set `allow_unknown_preds_category=True` to map them to void."
                )
        batch_preds = preds.reshape(-1, *preds.shape[-3:]) if preds.dim() > 3 else preds.unsqueeze(0)
        batch_target = target.reshape(-1, *target.shape[-3:]) if target.dim() > 3 else target.unsqueeze(0)
        mod_stuffs = self.stuffs if self._modified else None
        for p, t in zip(batch_preds, batch_target):
            iou_sum, tp, fp, fn = _panoptic_quality_update_sample(p, t, self.things, self.stuffs, mod_stuffs)
            for cat, v in iou_sum.items():
                self.iou_sum[self._cat_to_idx[cat]] += v
            for cat, v in tp.items():
                self.true_positives[self._cat_to_idx[cat]] += v
            for cat, v in fp.items():
                self.false_positives[self._cat_to_idx[cat]] += v
            for cat, v in fn.items():
                self.false_negatives[self._cat_to_idx[cat]] += v

    def compute(self) -> Tensor:
        """PQ (optionally SQ and RQ / per class)."""
        denom = self.true_positives + 0.5 * self.false_positives + 0.5 * self.false_negatives
        valid = denom > 0
        pq_per = torch.where(valid, self.iou_sum / denom.clamp(min=1e-9), torch.zeros_like(self.iou_sum))
        sq_per = torch.where(
            self.true_positives > 0, self.iou_sum / self.true_positives.clamp(min=1), torch.zeros_like(self.iou_sum)
        )
        rq_per = torch.where(valid, self.true_positives / denom.clamp(min=1e-9), torch.zeros_like(self.iou_sum))

        if self.return_per_class:
            # reference shapes (panoptic_qualities.py:223-226): (C, 3) with
            # sq_and_rq, else (1, C)
            if self.return_sq_and_rq:
                return torch.stack([pq_per, sq_per, rq_per], dim=-1)
            return pq_per.view(1, -1)
        pq = pq_per[valid].mean() if valid.any() else torch.tensor(0.0, dtype=torch.double)
        if self.return_sq_and_rq:
            sq = sq_per[valid].mean() if valid.any() else torch.tensor(0.0, dtype=torch.double)
            rq = rq_per[valid].mean() if valid.any() else torch.tensor(0.0, dtype=torch.double)
            return torch.stack([pq, sq, rq])
        return pq

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class ModifiedPanopticQuality(PanopticQuality):
    """Modified PQ (Porzi et al. 2019): stuff scored by per-pixel IoU, no 0.5 gate."""

    _modified = True


def panoptic_quality(
    preds: Tensor,
    target: Tensor,
    things: Collection[int],
    stuffs: Collection[int],
    allow_unknown_preds_category: bool = False,
    return_sq_and_rq: bool = False,
    return_per_class: bool = False,
) -> Tensor:
    """Functional PQ."""
    m = PanopticQuality(things, stuffs, allow_unknown_preds_category, return_sq_and_rq, return_per_class)
    m.update(preds, target)
    return m.compute()


def modified_panoptic_quality(
    preds: Tensor,
    target: Tensor,
    things: Collection[int],
    stuffs: Collection[int],
    allow_unknown_preds_category: bool = False,
) -> Tensor:
    """Functional modified PQ."""
    m = ModifiedPanopticQuality(things, stuffs, allow_unknown_preds_category)
    m.update(preds, target)
    return m.compute()
