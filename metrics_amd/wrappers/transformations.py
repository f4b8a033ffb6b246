"""Input-transformation wrappers. Parity: torchmetrics ``wrappers/transformations.py``."""
from __future__ import annotations

from typing import Any, Callable, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


class MetricInputTransformer(WrapperMetric):
    """Base: transform (preds, target) before delegating to the wrapped metric."""

    def __init__(self, wrapped_metric: Metric, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(wrapped_metric, Metric):
            raise TypeError(f"Expected wrapped metric to be an instance of `metrics_amd.Metric` but received {wrapped_metric}")
        self.wrapped_metric = wrapped_metric

    def transform_pred(self, pred: Tensor) -> Tensor:
        """Identity by default."""
        return pred

    def transform_target(self, target: Tensor) -> Tensor:
        """Identity by default."""
        return target

    def _wrap_transform(self, *args: Tensor) -> tuple:
        """Transform the positional (preds, target) args; pass any extras through."""
        out = list(args)
        if len(out) > 0:
            out[0] = self.transform_pred(out[0])
        if len(out) > 1:
            out[1] = self.transform_target(out[1])
        return tuple(out)

    def update(self, *args: Tensor, **kwargs: Any) -> None:
        """Transform inputs then update the wrapped metric (extra kwargs pass through)."""
        self.wrapped_metric.update(*self._wrap_transform(*args), **kwargs)

    def compute(self) -> Any:
        """Delegate."""
        return self.wrapped_metric.compute()

    def forward(self, *args: Tensor, **kwargs: Any) -> Any:
        self._forward_cache = self.wrapped_metric(*self._wrap_transform(*args), **kwargs)
        return self._forward_cache

    def reset(self) -> None:
        self.wrapped_metric.reset()
        super().reset()


class LambdaInputTransformer(MetricInputTransformer):
    """Apply user lambdas to preds/target before the wrapped metric."""

    def __init__(
        self,
        wrapped_metric: Metric,
        transform_pred: Optional[Callable] = None,
        transform_target: Optional[Callable] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(wrapped_metric, **kwargs)
        if transform_pred is not None and not callable(transform_pred):
            raise TypeError(f"Expected `transform_pred` to be a Callable but received {transform_pred}")
        if transform_target is not None and not callable(transform_target):
            raise TypeError(f"Expected `transform_target` to be a Callable but received {transform_target}")
        self._transform_pred_fn = transform_pred
        self._transform_target_fn = transform_target

    def transform_pred(self, pred: Tensor) -> Tensor:
        return self._transform_pred_fn(pred) if self._transform_pred_fn is not None else pred

    def transform_target(self, target: Tensor) -> Tensor:
        return self._transform_target_fn(target) if self._transform_target_fn is not None else target


class BinaryTargetTransformer(MetricInputTransformer):
    """Binarize targets at a threshold before the wrapped metric."""

    def __init__(self, wrapped_metric: Metric, threshold: float = 0, **kwargs: Any) -> None:
        super().__init__(wrapped_metric, **kwargs)
        if not isinstance(threshold, (int, float)):
            raise TypeError(f"Expected `threshold` to be a numeric value but received {threshold}")
        self.threshold = threshold

    def transform_target(self, target: Tensor) -> Tensor:
        return target.gt(self.threshold).long()
