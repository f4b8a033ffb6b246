"""Modular CosineSimilarity. Parity: torchmetrics ``regression/cosine_similarity.py``."""
from __future__ import annotations

from typing import Any, List, Optional

from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.functional.regression.cosine_similarity import _cosine_similarity_compute, _cosine_similarity_update


class CosineSimilarity(Metric):
    """Cosine similarity (stateful; cat state)."""

    is_differentiable = True
    higher_is_better = True
    full_state_update: bool = False
    plot_lower_bound: float = -1.0
    plot_upper_bound: float = 1.0

    preds: List[Tensor]
    target: List[Tensor]

    def __init__(self, reduction: Optional[str] = "sum", **kwargs: Any) -> None:
        super().__init__(**kwargs)
        allowed_reduction = ("sum", "mean", "none", None)
        if reduction not in allowed_reduction:
            raise ValueError(f"Expected argument `reduction` to be one of {allowed_reduction} but got {reduction}")
        self.reduction = reduction

        self.add_state("preds", [], dist_reduce_fx="cat")
        self.add_state("target", [], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Append the batch."""
        preds, target = _cosine_similarity_update(preds, target)
        self.preds.append(preds)
        self.target.append(target)

    def compute(self) -> Tensor:
        """Cosine similarity with the configured reduction."""
        preds = dim_zero_cat(self.preds)
        target = dim_zero_cat(self.target)
        return _cosine_similarity_compute(preds, target, self.reduction)

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)
