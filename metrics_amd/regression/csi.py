"""Modular Critical success index. Parity: torchmetrics ``regression/csi.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.regression.csi import _critical_success_index_compute, _critical_success_index_update


class CriticalSuccessIndex(Metric):
    """Critical success index (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    hits: Tensor
    misses: Tensor
    false_alarms: Tensor

    def __init__(self, threshold: float, keep_sequence_dim: Optional[int] = None, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.threshold = float(threshold)
        if keep_sequence_dim is not None and (not isinstance(keep_sequence_dim, int) or keep_sequence_dim < 0):
            raise ValueError(f"Expected keep_sequence_dim to be non-negative integer but got {keep_sequence_dim}")
        self.keep_sequence_dim = keep_sequence_dim

        if keep_sequence_dim is None:
            self.add_state("hits", default=torch.tensor(0), dist_reduce_fx="sum")
            self.add_state("misses", default=torch.tensor(0), dist_reduce_fx="sum")
            self.add_state("false_alarms", default=torch.tensor(0), dist_reduce_fx="sum")
        else:
            self.add_state("hits", default=[], dist_reduce_fx="cat")
            self.add_state("misses", default=[], dist_reduce_fx="cat")
            self.add_state("false_alarms", default=[], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate hit/miss/false-alarm counts."""
        hits, misses, false_alarms = _critical_success_index_update(
            preds, target, self.threshold, self.keep_sequence_dim
        )
        if self.keep_sequence_dim is None:
            self.hits = self.hits + hits
            self.misses = self.misses + misses
            self.false_alarms = self.false_alarms + false_alarms
        else:
            self.hits.append(hits)
            self.misses.append(misses)
            self.false_alarms.append(false_alarms)

    def compute(self) -> Tensor:
        from metrics_amd.utilities.data import dim_zero_cat

        if self.keep_sequence_dim is None:
            hits, misses, fa = self.hits, self.misses, self.false_alarms
        else:
            hits, misses, fa = dim_zero_cat(self.hits), dim_zero_cat(self.misses), dim_zero_cat(self.false_alarms)
        return _critical_success_index_compute(hits, misses, fa)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
