"""Kendall rank correlation (tau-a / tau-b / tau-c).

Parity: torchmetrics ``functional/regression/kendall.py`` (which counts pairs
with a per-element O(n^2) python loop). Here pair counting uses Knight's
O(n log n) algorithm: lexsort by (x, y), count y-inversions with a
divide-and-conquer merge (vectorized ``searchsorted`` per level), and derive
concordant/tie counts from group sizes.
"""
from __future__ import annotations

from typing import Optional, Tuple

import math

import numpy as np
import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape

_BASE = 64


def _inversions(a: np.ndarray) -> Tuple[int, np.ndarray]:
    """Count strict inversions (a[i] > a[j] for i < j); returns (count, sorted)."""
    n = a.size
    if n <= _BASE:
        if n < 2:
            return 0, a
        inv = int(np.sum(np.triu(a[:, None] > a[None, :], k=1)))
        return inv, np.sort(a, kind="stable")
    mid = n // 2
    inv_l, left = _inversions(a[:mid])
    inv_r, right = _inversions(a[mid:])
    # cross inversions: for each element of the right half, how many left
    # elements are strictly greater
    cross = left.size * right.size - int(np.searchsorted(left, right, side="right").sum())
    return inv_l + inv_r + cross, np.sort(np.concatenate([left, right]), kind="stable")


def _tie_group_sizes(sorted_keys: np.ndarray) -> np.ndarray:
    if sorted_keys.size == 0:
        return np.zeros(0, dtype=np.int64)
    change = np.nonzero(sorted_keys[1:] != sorted_keys[:-1])[0] + 1
    idx = np.concatenate([[0], change, [sorted_keys.size]])
    return np.diff(idx).astype(np.int64)


def _tie_pairs(sorted_keys: np.ndarray) -> int:
    """Sum of t*(t-1)/2 over runs of equal values in a sorted array."""
    t = _tie_group_sizes(sorted_keys)
    return int((t * (t - 1) // 2).sum())


def _tie_stats(sorted_keys: np.ndarray):
    """(pairs, p1, p2) tie statistics for the tau-b significance test:
    pairs = sum t(t-1)/2, p1 = sum t(t-1)(t-2), p2 = sum t(t-1)(2t+5)."""
    t = _tie_group_sizes(sorted_keys)
    pairs = float((t * (t - 1) // 2).sum())
    p1 = float((t * (t - 1) * (t - 2)).sum())
    p2 = float((t * (t - 1) * (2 * t + 5)).sum())
    return pairs, p1, p2


def _tie_pairs2(x: np.ndarray, y: np.ndarray) -> int:
    """Pairs tied in BOTH keys; x must be the primary sort key, y secondary."""
    if x.size < 2:
        return 0
    change = np.nonzero((x[1:] != x[:-1]) | (y[1:] != y[:-1]))[0] + 1
    idx = np.concatenate([[0], change, [x.size]])
    t = np.diff(idx).astype(np.int64)
    return int((t * (t - 1) // 2).sum())


def _count_pairs(x: Tensor, y: Tensor):
    """(concordant, discordant, ties_x, ties_y, tie-stat tuples) via Knight's algorithm."""
    xn = x.detach().cpu().numpy().astype(np.float64)
    yn = y.detach().cpu().numpy().astype(np.float64)
    n = xn.size
    order = np.lexsort((yn, xn))  # sort by x, ties broken by y ascending
    xs, ys = xn[order], yn[order]
    tot = n * (n - 1) // 2
    x_stats = _tie_stats(xs)
    y_stats = _tie_stats(np.sort(yn, kind="stable"))
    xtie = x_stats[0]
    ytie = y_stats[0]
    xytie = _tie_pairs2(xs, ys)
    dis, _ = _inversions(ys)
    both_untied = tot - xtie - ytie + xytie
    con = both_untied - dis

    dev = x.device
    as_t = lambda v: torch.tensor(float(v), dtype=torch.float64, device=dev)
    return as_t(con), as_t(dis), as_t(xtie), as_t(ytie), x_stats, y_stats


def _kendall_corrcoef_compute(
    preds: Tensor, target: Tensor, variant: str = "b", alternative: Optional[str] = None
) -> Tuple[Tensor, Optional[Tensor]]:
    if preds.ndim == 1:
        preds = preds.unsqueeze(1)
        target = target.unsqueeze(1)
    taus = []
    p_values = []
    for d in range(preds.shape[1]):
        x = preds[:, d].double()
        y = target[:, d].double()
        n = x.numel()
        con, dis, tie_x, tie_y, x_stats, y_stats = _count_pairs(x, y)
        n0 = n * (n - 1) / 2
        if variant == "a":
            # reference tau-a: (con - dis) / (con + dis)  (ties excluded)
            tau = (con - dis) / (con + dis)
        elif variant == "b":
            tau = (con - dis) / torch.sqrt((n0 - tie_x) * (n0 - tie_y))
        elif variant == "c":
            m = min(len(torch.unique(x)), len(torch.unique(y)))
            tau = 2 * (con - dis) / (n**2 * (m - 1) / m)
        else:
            raise ValueError(f"Unknown variant {variant}")
        taus.append(tau.float())
        if alternative is not None:
            # reference tie-corrected significance test
            # (reference functional/regression/kendall.py:192-222)
            cmd = float(con - dis)
            base = n * (n - 1) * (2 * n + 5)
            if variant == "a":
                t_value = 3 * cmd / math.sqrt(base / 2)
            else:
                mm = n * (n - 1)
                denom = (base - x_stats[2] - y_stats[2]) / 18
                denom += (2 * x_stats[0] * y_stats[0]) / mm
                # n == 2 gives 0/0 here in the reference (always nan)
                denom += x_stats[1] * y_stats[1] / (9 * mm * (n - 2)) if n != 2 else float("nan")
                t_value = cmd / math.sqrt(denom) if denom == denom and denom > 0 else float("nan")
            p_values.append(t_value)
    tau_t = torch.stack(taus).squeeze()

    p_value = None
    if alternative is not None:
        if alternative not in ("two-sided", "greater", "less"):
            raise ValueError(f"Unknown alternative {alternative}")
        t = torch.tensor(p_values, dtype=torch.float32)
        if alternative == "two-sided":
            t = t.abs()
        if alternative in ("two-sided", "greater"):
            t = -t
        normal = torch.distributions.Normal(0.0, 1.0)
        is_nan = t.isnan()
        p_value = normal.cdf(t.nan_to_num())
        p_value = p_value.where(~is_nan, torch.tensor(float("nan")))
        if alternative == "two-sided":
            p_value = 2 * p_value
        p_value = p_value.squeeze()
    return tau_t, p_value


def kendall_rank_corrcoef(
    preds: Tensor,
    target: Tensor,
    variant: str = "b",
    t_test: bool = False,
    alternative: Optional[str] = "two-sided",
):
    """Kendall rank correlation coefficient (optionally with p-value)."""
    _check_same_shape(preds, target)
    if t_test:
        tau, p = _kendall_corrcoef_compute(preds, target, variant, alternative)
        return tau, p
    tau, _ = _kendall_corrcoef_compute(preds, target, variant, None)
    return tau
