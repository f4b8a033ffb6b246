"""Critical success index. Parity: torchmetrics ``functional/regression/csi.py``."""
from __future__ import annotations

from typing import Tuple

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape


def _critical_success_index_update(
    preds: Tensor, target: Tensor, threshold: float, keep_sequence_dim: int = None
) -> Tuple[Tensor, Tensor, Tensor]:
    """Binarize at ``threshold`` and count hits / misses / false alarms."""
    _check_same_shape(preds, target)
    if keep_sequence_dim is None:
        sum_dims = None
    elif not 0 <= keep_sequence_dim < preds.ndim:
        raise ValueError(f"Expected keep_sequence_dim to be in range [0, {preds.ndim}) but got {keep_sequence_dim}")
    else:
        sum_dims = tuple(i for i in range(preds.ndim) if i != keep_sequence_dim)

    preds_bin = preds >= threshold
    target_bin = target >= threshold

    if sum_dims is None:
        hits = (preds_bin & target_bin).sum()
        misses = (~preds_bin & target_bin).sum()
        false_alarms = (preds_bin & ~target_bin).sum()
    else:
        hits = (preds_bin & target_bin).sum(dim=sum_dims)
        misses = (~preds_bin & target_bin).sum(dim=sum_dims)
        false_alarms = (preds_bin & ~target_bin).sum(dim=sum_dims)
    return hits, misses, false_alarms


def _critical_success_index_compute(hits: Tensor, misses: Tensor, false_alarms: Tensor) -> Tensor:
    return hits / (hits + misses + false_alarms)


def critical_success_index(
    preds: Tensor, target: Tensor, threshold: float, keep_sequence_dim: int = None
) -> Tensor:
    """Critical success index (threat score)."""
    hits, misses, false_alarms = _critical_success_index_update(preds, target, threshold, keep_sequence_dim)
    return _critical_success_index_compute(hits, misses, false_alarms)
