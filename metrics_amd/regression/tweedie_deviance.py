"""Modular Tweedie deviance score. Parity: torchmetrics ``regression/tweedie_deviance.py``."""
from __future__ import annotations

from typing import Any, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.regression.tweedie_deviance import _tweedie_deviance_score_compute, _tweedie_deviance_score_update


class TweedieDevianceScore(Metric):
    """Tweedie deviance score (stateful)."""

    is_differentiable = True
    higher_is_better = False
    full_state_update = False
    plot_lower_bound: float = 0.0

    sum_deviance_score: Tensor
    num_observations: Tensor

    def __init__(self, power: float = 0.0, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if 0 < power < 1:
            raise ValueError(f"Deviance Score is not defined for power={power}.")
        self.power = power
        self.add_state("sum_deviance_score", default=torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("num_observations", default=torch.tensor(0), dist_reduce_fx="sum")

    def update(self, preds: Tensor, targets: Tensor) -> None:
        """Accumulate deviance sums."""
        sum_deviance_score, num_observations = _tweedie_deviance_score_update(preds, targets, self.power)
        self.sum_deviance_score = self.sum_deviance_score + sum_deviance_score
        self.num_observations = self.num_observations + num_observations

    def compute(self) -> Tensor:
        return _tweedie_deviance_score_compute(self.sum_deviance_score, self.num_observations)

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
