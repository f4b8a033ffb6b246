"""Running-window wrapper.

Parity: torchmetrics ``wrappers/running.py`` — keeps ``window`` copies of each
base-metric state (attributes ``{key}_{i}``), update writes the slot
``num_seen % window`` then resets the base metric; compute merges the window
via the base metric's ``_reduce_states``.
"""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Optional, Union

from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


class Running(WrapperMetric):
    """Compute any metric over the last ``window`` update calls."""

    def __init__(self, base_metric: Metric, window: int = 5) -> None:
        super().__init__()
        if not isinstance(base_metric, Metric):
            raise ValueError(
                f"Expected argument `metric` to be an instance of `metrics_amd.Metric` but got {base_metric}"
            )
        if not (isinstance(window, int) and window > 0):
            raise ValueError(f"Expected argument `window` to be a positive integer but got {window}")
        self.base_metric = base_metric
        self.window = window

        if base_metric.full_state_update is not False:
            raise ValueError(
                f"Expected attribute `full_state_update` set to `False` but got {base_metric.full_state_update}"
            )
        self._num_vals_seen = 0

        for key in base_metric._defaults:
            for i in range(window):
                self.add_state(
                    name=key + f"_{i}", default=deepcopy(base_metric._defaults[key]),
                    dist_reduce_fx=base_metric._reductions[key],
                )

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Update the rolling window states."""
        val = self._num_vals_seen % self.window
        self.base_metric.update(*args, **kwargs)
        for key in self.base_metric._defaults:
            setattr(self, key + f"_{val}", deepcopy(getattr(self.base_metric, key)))
        self.base_metric.reset()
        self._num_vals_seen += 1

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        """Update the window with this batch and return THIS batch's value.

        The running value is what ``compute()`` returns (reference
        wrappers/running.py:116-125 forwards the base metric and snapshots its
        states into the window slot).
        """
        val = self._num_vals_seen % self.window
        res = self.base_metric.forward(*args, **kwargs)
        for key in self.base_metric._defaults:
            setattr(self, key + f"_{val}", deepcopy(getattr(self.base_metric, key)))
        self.base_metric.reset()
        self._num_vals_seen += 1
        self._computed = None
        self._forward_cache = res
        return res

    def compute(self) -> Any:
        """Merge the window states into the base metric and compute."""
        for i in range(self.window):
            self.base_metric._reduce_states(
                {key: getattr(self, key + f"_{i}") for key in self.base_metric._defaults}
            )
        self.base_metric._update_count = self._num_vals_seen
        val = self.base_metric.compute()
        self.base_metric.reset()
        return val

    def reset(self) -> None:
        super().reset()
        self.base_metric.reset()
        self._num_vals_seen = 0

    def plot(self, val: Optional[Union[Tensor, Any]] = None, ax: Optional[Any] = None) -> Any:
        return self._plot(val, ax)
