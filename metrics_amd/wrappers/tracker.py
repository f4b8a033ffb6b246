"""MetricTracker. Parity: torchmetrics ``wrappers/tracker.py``."""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Dict, List, Optional, Tuple, Union

import torch
from torch import Tensor
from torch.nn import ModuleList

from metrics_amd.collections import MetricCollection
from metrics_amd.metric import Metric
from metrics_amd.utilities.prints import rank_zero_warn


class MetricTracker(ModuleList):
    """Track a metric (or collection) over time steps; query best values.

    ``increment()`` starts a new step (a fresh copy of the base metric);
    update/forward/compute hit the latest step; ``compute_all()`` /
    ``best_metric()`` summarize history.
    """

    def __init__(self, metric: Union[Metric, MetricCollection], maximize: Union[bool, List[bool]] = True) -> None:
        super().__init__()
        if not isinstance(metric, (Metric, MetricCollection)):
            raise TypeError(
                "Metric arg need to be an instance of a metrics_amd"
                f" `Metric` or `MetricCollection` but got {metric}"
            )
        self._base_metric = metric
        if not isinstance(maximize, (bool, list)):
            raise ValueError("Argument `maximize` should either be a single bool or list of bool")
        if isinstance(maximize, list) and not all(isinstance(m, bool) for m in maximize):
            raise ValueError("Argument `maximize` should either be a single bool or list of bool")
        if isinstance(maximize, list) and isinstance(metric, MetricCollection) and len(maximize) != len(metric):
            raise ValueError("The len of argument `maximize` should match the length of the metric collection")
        if isinstance(metric, Metric) and not isinstance(maximize, bool):
            raise ValueError("Argument `maximize` should be a single bool when `metric` is a single Metric")
        self.maximize = maximize

        self._increment_called = False

    @property
    def n_steps(self) -> int:
        """Number of steps tracked so far."""
        return max(len(self) - 1, 0)  # subtract the base metric

    def increment(self) -> None:
        """Start a new tracking step."""
        self._increment_called = True
        self.append(deepcopy(self._base_metric))

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        """Forward on the latest step's metric."""
        self._check_for_increment("forward")
        return self[-1](*args, **kwargs)

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Update the latest step's metric."""
        self._check_for_increment("update")
        self[-1].update(*args, **kwargs)

    def compute(self) -> Any:
        """Compute the latest step's metric."""
        self._check_for_increment("compute")
        return self[-1].compute()

    def compute_all(self) -> Any:
        """Compute every tracked step."""
        self._check_for_increment("compute_all")
        # The i=0 is the base metric prototype — skip it
        res = [metric.compute() for i, metric in enumerate(self) if i != 0]
        try:
            if isinstance(res[0], dict):
                keys = res[0].keys()
                return {k: torch.stack([r[k] for r in res], dim=0) for k in keys}
            if isinstance(res[0], list):
                return torch.stack([torch.stack(r, dim=0) for r in res], 0)
            return torch.stack(res, dim=0)
        except TypeError:
            return res

    def reset(self) -> None:
        """Reset the latest step's metric."""
        self[-1].reset()

    def reset_all(self) -> None:
        """Reset every tracked step."""
        for metric in self:
            metric.reset()

    def plot(self, val=None, ax=None):
        """Plot the tracked metric values over increments (reference wrappers/tracker.py:305)."""
        from metrics_amd.utilities.plot import plot_single_or_multi_val

        val = val if val is not None else self.compute_all()
        if isinstance(val, dict):
            import matplotlib.pyplot as plt

            fig, ax_ = (plt.subplots() if ax is None else (None, ax))
            for k, v in val.items():
                plot_single_or_multi_val(list(v) if v.ndim else v, ax=ax_)
            return fig, ax_
        if isinstance(val, (list, tuple)):
            return plot_single_or_multi_val(list(val), ax=ax)
        return plot_single_or_multi_val(list(val.flatten()) if val.ndim else val, ax=ax)

    def best_metric(
        self, return_step: bool = False
    ) -> Union[
        None, float, Tuple[float, int], Tuple[None, None], Dict[str, Union[float, None]],
        Tuple[Dict[str, Union[float, None]], Dict[str, Union[int, None]]],
    ]:
        """Best value (and optionally step) over history."""
        res = self.compute_all()
        if isinstance(res, list):
            rank_zero_warn(
                "Encountered nested data structures that could not be stacked; `best_metric` returns None",
                UserWarning,
            )
            return (None, None) if return_step else None

        if isinstance(self._base_metric, Metric):
            fn = torch.max if self.maximize else torch.min
            try:
                value, idx = fn(res, 0)
                if return_step:
                    return value.item(), idx.item()
                return value.item()
            except (ValueError, RuntimeError) as error:
                rank_zero_warn(
                    f"Encountered the following error when trying to get the best metric: {error}"
                    "this is probably due to the 'best' not being defined for this metric."
                    "Returning `None` instead.",
                    UserWarning,
                )
                return (None, None) if return_step else None
        else:
            maximize = self.maximize if isinstance(self.maximize, list) else len(res) * [self.maximize]
            value, idx = {}, {}
            for i, (k, v) in enumerate(res.items()):
                try:
                    fn = torch.max if maximize[i] else torch.min
                    out = fn(v, 0)
                    value[k], idx[k] = out[0].item(), out[1].item()
                except (ValueError, RuntimeError) as error:
                    rank_zero_warn(
                        f"Encountered the following error when trying to get the best metric for metric {k}:"
                        f"{error} this is probably due to the 'best' not being defined for this metric."
                        "Returning `None` instead.",
                        UserWarning,
                    )
                    value[k], idx[k] = None, None
            if return_step:
                return value, idx
            return value

    def _check_for_increment(self, method: str) -> None:
        if not self._increment_called:
            raise ValueError(f"`{method}` cannot be called before `.increment()` has been called.")
