"""MultitaskWrapper. Parity: torchmetrics ``wrappers/multitask.py``."""
from __future__ import annotations

from typing import Any, Dict, Optional, Union

from torch import Tensor
from torch.nn import ModuleDict

from metrics_amd.collections import MetricCollection
from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


class MultitaskWrapper(WrapperMetric):
    """Route per-task (preds, target) dicts to per-task metrics."""

    is_differentiable = False

    def __init__(
        self,
        task_metrics: Dict[str, Union[Metric, MetricCollection]],
        prefix: Optional[str] = None,
        postfix: Optional[str] = None,
    ) -> None:
        super().__init__()
        if not isinstance(task_metrics, dict):
            raise TypeError(f"Expected argument `task_metrics` to be a dict. Found task_metrics = {task_metrics}")
        for metric in task_metrics.values():
            if not (isinstance(metric, (Metric, MetricCollection))):
                raise TypeError(
                    "Expected each task's metric to be a Metric or a MetricCollection. "
                    f"Found a metric of type {type(metric)}"
                )
        self.task_metrics = ModuleDict(task_metrics)
        if prefix is not None and not isinstance(prefix, str):
            raise ValueError(f"Expected argument `prefix` to either be `None` or a string but got {prefix}")
        self._prefix = prefix or ""
        if postfix is not None and not isinstance(postfix, str):
            raise ValueError(f"Expected argument `postfix` to either be `None` or a string but got {postfix}")
        self._postfix = postfix or ""

    def items(self, flatten: bool = True):
        """(task_name, metric) pairs; flattens collections when ``flatten``."""
        for task_name, metric in self.task_metrics.items():
            if flatten and isinstance(metric, MetricCollection):
                for sub_name, sub_metric in metric.items():
                    yield f"{task_name}_{sub_name}", sub_metric
            else:
                yield task_name, metric

    def keys(self, flatten: bool = True):
        for name, _ in self.items(flatten):
            yield name

    def values(self, flatten: bool = True):
        for _, metric in self.items(flatten):
            yield metric

    def update(self, task_preds: Dict[str, Any], task_targets: Dict[str, Any]) -> None:
        """Update each task metric with its own preds/target."""
        if not self.task_metrics.keys() == task_preds.keys() == task_targets.keys():
            raise ValueError(
                "Expected arguments `task_preds` and `task_targets` to have the same keys as the wrapped `task_metrics`. "
                f"Found task_preds.keys() = {task_preds.keys()}, task_targets.keys() = {task_targets.keys()} "
                f"and self.task_metrics.keys() = {self.task_metrics.keys()}"
            )
        for task_name, metric in self.task_metrics.items():
            metric.update(task_preds[task_name], task_targets[task_name])

    def compute(self) -> Dict[str, Any]:
        """Per-task results dict."""
        return {self._prefix + k + self._postfix: m.compute() for k, m in self.task_metrics.items()}

    def forward(self, task_preds: Dict[str, Any], task_targets: Dict[str, Any]) -> Dict[str, Any]:
        return {
            self._prefix + task_name + self._postfix: metric(task_preds[task_name], task_targets[task_name])
            for task_name, metric in self.task_metrics.items()
        }

    def reset(self) -> None:
        for metric in self.task_metrics.values():
            metric.reset()
        super().reset()

    def clone(self, prefix: Optional[str] = None, postfix: Optional[str] = None) -> "MultitaskWrapper":
        """Deep copy with optional re-keying."""
        from copy import deepcopy

        mt = deepcopy(self)
        if prefix is not None:
            mt._prefix = prefix
        if postfix is not None:
            mt._postfix = postfix
        return mt

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
