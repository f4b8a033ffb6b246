"""Minkowski distance. Parity: torchmetrics ``functional/regression/minkowski.py``."""
from __future__ import annotations

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape
from metrics_amd.utilities.exceptions import MetricsUserError


def _minkowski_distance_update(preds: Tensor, targets: Tensor, p: float) -> Tensor:
    _check_same_shape(preds, targets)
    if not (isinstance(p, (float, int)) and p >= 1):
        raise MetricsUserError(f"Argument ``p`` must be a float or int greater than 1, but got {p}")
    difference = torch.abs(preds - targets)
    return torch.sum(torch.pow(difference, p))


def _minkowski_distance_compute(distance: Tensor, p: float) -> Tensor:
    return torch.pow(distance, 1.0 / p)


def minkowski_distance(preds: Tensor, targets: Tensor, p: float) -> Tensor:
    """Minkowski distance of order p."""
    minkowski_dist_sum = _minkowski_distance_update(preds, targets, p)
    return _minkowski_distance_compute(minkowski_dist_sum, p)
