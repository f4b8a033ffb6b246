"""L5 — aggregation metrics.

Parity: torchmetrics ``aggregation.py`` (BaseAggregator, MaxMetric, MinMetric,
SumMetric, CatMetric, MeanMetric, RunningMean, RunningSum).
"""
from __future__ import annotations

from typing import Any, Callable, List, Optional, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.utilities.exceptions import MetricsUserError
from metrics_amd.utilities.prints import rank_zero_warn


class BaseAggregator(Metric):
    """Base class for aggregation of a stream of values (handles nan strategies)."""

    is_differentiable = None
    higher_is_better = None
    full_state_update: bool = False

    def __init__(
        self,
        fn: Union[Callable, str],
        default_value: Union[Tensor, List],
        nan_strategy: Union[str, float] = "error",
        state_name: str = "value",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        allowed_nan_strategy = ("error", "warn", "ignore", "disable")
        if nan_strategy not in allowed_nan_strategy and not isinstance(nan_strategy, float):
            raise ValueError(
                f"Arg `nan_strategy` should either be a float or one of {allowed_nan_strategy} but got {nan_strategy}."
            )

        self.nan_strategy = nan_strategy
        self.add_state(state_name, default=default_value, dist_reduce_fx=fn)
        self.state_name = state_name

    def _cast_and_nan_check_input(
        self, x: Union[float, Tensor], weight: Optional[Union[float, Tensor]] = None
    ) -> tuple:
        """Cast input to tensor and replace/raise on nans according to the strategy."""
        if not isinstance(x, Tensor):
            x = torch.as_tensor(x, dtype=self.dtype, device=self.device)
        if weight is not None and not isinstance(weight, Tensor):
            weight = torch.as_tensor(weight, dtype=self.dtype, device=self.device)

        if weight is None:
            weight = torch.ones_like(x)
            weight_was_expanded = False
        else:
            # scalar/broadcastable weights line up elementwise with the values
            weight_was_expanded = weight.shape != x.shape
            weight = torch.broadcast_to(weight, x.shape)

        if self.nan_strategy != "disable":
            nans = torch.isnan(x)
            nans_weight = torch.isnan(weight)
            anynan = nans.any() or nans_weight.any()
            if anynan:
                if self.nan_strategy == "error":
                    raise RuntimeError("Encountered `nan` values in tensor")
                if self.nan_strategy in ("ignore", "warn"):
                    if self.nan_strategy == "warn":
                        rank_zero_warn("Encountered `nan` values in tensor. Will be removed.", UserWarning)
                    x = x[~(nans | nans_weight)]
                    weight = weight[~(nans | nans_weight)]
                else:  # float strategy
                    if not isinstance(self.nan_strategy, float):
                        raise ValueError(f"`nan_strategy` shall be float but you pass {self.nan_strategy}")
                    mask = nans | nans_weight
                    x = x.clone()
                    x[mask] = self.nan_strategy
                    if weight_was_expanded:
                        # reference parity: it writes through the broadcast
                        # VIEW, so every aliased element of an expanded weight
                        # takes the fill value — any nan floods the weight
                        weight = torch.full_like(x, self.nan_strategy)
                    else:
                        weight = weight.clone()
                        weight[mask] = self.nan_strategy

        return x.to(self.dtype), weight.to(self.dtype)

    def update(self, value: Union[float, Tensor]) -> None:
        """Override in child class."""

    def compute(self) -> Tensor:
        """Compute the aggregated value."""
        return getattr(self, self.state_name)


class MaxMetric(BaseAggregator):
    """Running maximum of a stream of values."""

    full_state_update: bool = True
    plot_lower_bound = None

    def __init__(self, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(
            "max",
            -torch.tensor(float("inf")),
            nan_strategy,
            state_name="max_value",
            **kwargs,
        )

    def update(self, value: Union[float, Tensor]) -> None:
        value, _ = self._cast_and_nan_check_input(value)
        if value.numel():
            self.max_value = torch.max(self.max_value, torch.max(value))

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MinMetric(BaseAggregator):
    """Running minimum of a stream of values."""

    full_state_update: bool = True

    def __init__(self, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(
            "min",
            torch.tensor(float("inf")),
            nan_strategy,
            state_name="min_value",
            **kwargs,
        )

    def update(self, value: Union[float, Tensor]) -> None:
        value, _ = self._cast_and_nan_check_input(value)
        if value.numel():
            self.min_value = torch.min(self.min_value, torch.min(value))

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class SumMetric(BaseAggregator):
    """Running sum of a stream of values."""

    def __init__(self, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(
            "sum",
            torch.tensor(0.0, dtype=torch.get_default_dtype()),
            nan_strategy,
            state_name="sum_value",
            **kwargs,
        )

    def update(self, value: Union[float, Tensor]) -> None:
        value, _ = self._cast_and_nan_check_input(value)
        if value.numel():
            self.sum_value += value.sum()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class CatMetric(BaseAggregator):
    """Concatenate a stream of values."""

    def __init__(self, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__("cat", [], nan_strategy, **kwargs)

    def update(self, value: Union[float, Tensor]) -> None:
        value, _ = self._cast_and_nan_check_input(value)
        if value.numel():
            self.value.append(value)

    def compute(self) -> Tensor:
        if isinstance(self.value, list) and self.value:
            return dim_zero_cat(self.value)
        return self.value


class MeanMetric(BaseAggregator):
    """(Weighted) running mean of a stream of values.

    States ``mean_value`` (weighted sum) and ``weight`` both sync with a
    fused RCCL all-reduce.
    """

    def __init__(self, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(
            "sum",
            torch.tensor(0.0, dtype=torch.get_default_dtype()),
            nan_strategy,
            state_name="mean_value",
            **kwargs,
        )
        self.add_state("weight", default=torch.tensor(0.0, dtype=torch.get_default_dtype()), dist_reduce_fx="sum")

    def update(self, value: Union[float, Tensor], weight: Union[float, Tensor] = 1.0) -> None:
        """Accumulate ``value`` with element-wise ``weight`` (broadcastable)."""
        value, weight = self._cast_and_nan_check_input(value, weight)
        if value.numel() == 0:
            return
        # expand the weight so it lines up elementwise with value
        weight = torch.broadcast_to(weight, value.shape)
        self.mean_value += (value * weight).sum()
        self.weight += weight.sum()

    def compute(self) -> Tensor:
        return self.mean_value / self.weight

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


# Running window aggregations live with the Running wrapper:
from metrics_amd.wrappers.running import Running  # noqa: E402


class RunningMean(Running):
    """Mean over the last ``window`` updates."""

    def __init__(self, window: int = 5, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(base_metric=MeanMetric(nan_strategy=nan_strategy, **kwargs), window=window)


class RunningSum(Running):
    """Sum over the last ``window`` updates."""

    def __init__(self, window: int = 5, nan_strategy: Union[str, float] = "warn", **kwargs: Any) -> None:
        super().__init__(base_metric=SumMetric(nan_strategy=nan_strategy, **kwargs), window=window)


__all__ = [
    "BaseAggregator",
    "CatMetric",
    "MaxMetric",
    "MeanMetric",
    "MinMetric",
    "RunningMean",
    "RunningSum",
    "SumMetric",
]
