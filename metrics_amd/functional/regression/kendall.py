"""Kendall rank correlation (tau-a / tau-b / tau-c).

Parity: torchmetrics ``functional/regression/kendall.py``. Concordant/
discordant pair counting is done with a vectorized O(n^2/chunk) loop —
fine for the list-state sizes metrics see; a merge-sort O(n log n) kernel is
a future optimization.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape

_CHUNK = 2048


def _count_pairs(x: Tensor, y: Tensor) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Return (concordant, discordant, ties_x_only, ties_y_only) pair counts."""
    n = x.numel()
    con = torch.zeros((), dtype=torch.float64, device=x.device)
    dis = torch.zeros((), dtype=torch.float64, device=x.device)
    tie_x = torch.zeros((), dtype=torch.float64, device=x.device)
    tie_y = torch.zeros((), dtype=torch.float64, device=x.device)
    for i0 in range(0, n, _CHUNK):
        xi = x[i0 : i0 + _CHUNK].unsqueeze(1)
        yi = y[i0 : i0 + _CHUNK].unsqueeze(1)
        # only pairs (i, j) with j > i
        xj = x.unsqueeze(0)
        yj = y.unsqueeze(0)
        mask = torch.arange(n, device=x.device).unsqueeze(0) > (
            torch.arange(i0, min(i0 + _CHUNK, n), device=x.device).unsqueeze(1)
        )
        sx = torch.sign(xj - xi)
        sy = torch.sign(yj - yi)
        prod = sx * sy
        con += ((prod > 0) & mask).sum()
        dis += ((prod < 0) & mask).sum()
        tie_x += ((sx == 0) & (sy != 0) & mask).sum()
        tie_y += ((sy == 0) & (sx != 0) & mask).sum()
    return con, dis, tie_x, tie_y


def _kendall_corrcoef_compute(
    preds: Tensor, target: Tensor, variant: str = "b", alternative: Optional[str] = None
) -> Tuple[Tensor, Optional[Tensor]]:
    if preds.ndim == 1:
        preds = preds.unsqueeze(1)
        target = target.unsqueeze(1)
    taus = []
    for d in range(preds.shape[1]):
        x = preds[:, d].double()
        y = target[:, d].double()
        n = x.numel()
        con, dis, tie_x, tie_y = _count_pairs(x, y)
        n0 = n * (n - 1) / 2
        if variant == "a":
            tau = (con - dis) / n0
        elif variant == "b":
            # total ties (pairs tied in x, in y — incl. both)
            sx = x.unsqueeze(0) - x.unsqueeze(1)
            sy = y.unsqueeze(0) - y.unsqueeze(1)
            iu = torch.triu_indices(n, n, offset=1, device=x.device)
            tx = (sx[iu[0], iu[1]] == 0).sum().double()
            ty = (sy[iu[0], iu[1]] == 0).sum().double()
            tau = (con - dis) / torch.sqrt((n0 - tx) * (n0 - ty))
        elif variant == "c":
            m = min(len(torch.unique(x)), len(torch.unique(y)))
            tau = 2 * (con - dis) / (n**2 * (m - 1) / m)
        else:
            raise ValueError(f"Unknown variant {variant}")
        taus.append(tau.float())
    tau_t = torch.stack(taus).squeeze()

    p_value = None
    if alternative is not None:
        # normal approximation for the two-sided test
        n = preds.shape[0]
        var = torch.tensor(2.0 * (2 * n + 5) / (9 * n * (n - 1)), device=tau_t.device)
        z = tau_t / var.sqrt()
        normal = torch.distributions.Normal(0.0, 1.0)
        if alternative == "two-sided":
            p_value = 2 * (1 - normal.cdf(z.abs()))
        elif alternative == "greater":
            p_value = 1 - normal.cdf(z)
        elif alternative == "less":
            p_value = normal.cdf(z)
        else:
            raise ValueError(f"Unknown alternative {alternative}")
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
