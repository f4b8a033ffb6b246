"""Abstract base class for wrapper metrics.

Parity: torchmetrics ``wrappers/abstract.py``.
"""
from __future__ import annotations

import functools
from typing import Any, Callable

import torch

from metrics_amd.metric import Metric


class WrapperMetric(Metric):
    """Base class for metrics that wrap other metrics.

    Wrappers delegate synchronization and result caching to the wrapped
    metric(s): the wrapper's own ``compute`` is left unwrapped (no sync, no
    cache), and ``update`` only handles counting + grad-mode.
    """

    def _wrap_update(self, update: Callable) -> Callable:
        @functools.wraps(update)
        def wrapped_func(*args: Any, **kwargs: Any) -> None:
            self._computed = None
            self._update_count += 1
            with torch.set_grad_enabled(self._enable_grad):
                update(*args, **kwargs)

        return wrapped_func

    def _wrap_compute(self, compute: Callable) -> Callable:
        @functools.wraps(compute)
        def wrapped_func(*args: Any, **kwargs: Any) -> Any:
            return compute(*args, **kwargs)

        return wrapped_func

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        """Wrappers define their own forward; default mirrors update-then-compute."""
        self.update(*args, **kwargs)
        self._forward_cache = None
        return self._forward_cache
