"""Task-wrapper base for classification metrics.

Parity: torchmetrics ``classification/base.py`` — classes like ``Accuracy``
are constructors that dispatch on ``task`` and return the Binary/Multiclass/
Multilabel concrete class from ``__new__``.
"""
from __future__ import annotations

from typing import Any

from metrics_amd.metric import Metric


class _ClassificationTaskWrapper(Metric):
    """Base for wrapper metrics whose ``__new__`` returns a task-specific metric."""

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Not callable: instantiation dispatches to a task class."""
        raise NotImplementedError(f"{self.__class__.__name__} metric does not have an `update` method.")

    def compute(self) -> None:
        """Not callable: instantiation dispatches to a task class."""
        raise NotImplementedError(f"{self.__class__.__name__} metric does not have a `compute` method.")
