"""MultioutputWrapper. Parity: torchmetrics ``wrappers/multioutput.py``."""
from __future__ import annotations

from copy import deepcopy
from typing import Any, List, Optional, Tuple

import torch
from torch import Tensor
from torch.nn import ModuleList

from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


def _get_nan_indices(*tensors: Tensor) -> Tensor:
    """Rows where ANY tensor has a nan."""
    if len(tensors) == 0:
        raise ValueError("Must pass at least one tensor as argument")
    sentinel_shape = (-1,) + tensors[0].shape[1:]
    nan_idxs = torch.zeros(len(tensors[0]), dtype=torch.bool, device=tensors[0].device)
    for tensor in tensors:
        permuted_tensor = tensor.flatten(start_dim=1)
        nan_idxs |= torch.any(permuted_tensor.isnan(), dim=1)
    return nan_idxs


class MultioutputWrapper(WrapperMetric):
    """Apply a metric independently to each output dimension (last dim)."""

    is_differentiable = False

    def __init__(
        self,
        base_metric: Metric,
        num_outputs: int,
        output_dim: int = -1,
        remove_nans: bool = True,
        squeeze_outputs: bool = True,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.metrics = ModuleList([deepcopy(base_metric) for _ in range(num_outputs)])
        self.output_dim = output_dim
        self.remove_nans = remove_nans
        self.squeeze_outputs = squeeze_outputs

    def _get_args_kwargs_by_output(self, *args: Tensor, **kwargs: Tensor) -> List[Tuple]:
        """Slice inputs along the output dimension, per output."""
        args_kwargs_by_output = []
        for i in range(len(self.metrics)):
            selected_args = [
                torch.index_select(arg, self.output_dim, torch.tensor(i, device=arg.device)) for arg in args
            ]
            selected_kwargs = {
                k: torch.index_select(v, self.output_dim, torch.tensor(i, device=v.device)) for k, v in kwargs.items()
            }
            if self.remove_nans:
                tensors = selected_args + list(selected_kwargs.values())
                nan_idxs = _get_nan_indices(*tensors)
                selected_args = [arg[~nan_idxs] for arg in selected_args]
                selected_kwargs = {k: v[~nan_idxs] for k, v in selected_kwargs.items()}

            if self.squeeze_outputs:
                selected_args = [arg.squeeze(self.output_dim) for arg in selected_args]
                selected_kwargs = {k: v.squeeze(self.output_dim) for k, v in selected_kwargs.items()}
            args_kwargs_by_output.append((selected_args, selected_kwargs))
        return args_kwargs_by_output

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Update each per-output metric on its slice."""
        reshaped_args_kwargs = self._get_args_kwargs_by_output(*args, **kwargs)
        for metric, (selected_args, selected_kwargs) in zip(self.metrics, reshaped_args_kwargs):
            metric.update(*selected_args, **selected_kwargs)

    def compute(self) -> Tensor:
        """Stacked per-output results."""
        return torch.stack([m.compute() for m in self.metrics], 0)

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        reshaped_args_kwargs = self._get_args_kwargs_by_output(*args, **kwargs)
        results = [
            metric(*selected_args, **selected_kwargs)
            for metric, (selected_args, selected_kwargs) in zip(self.metrics, reshaped_args_kwargs)
        ]
        if results[0] is None:
            self._forward_cache = None
            return self._forward_cache
        self._forward_cache = torch.stack(results, 0)
        return self._forward_cache

    def reset(self) -> None:
        for metric in self.metrics:
            metric.reset()
        super().reset()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
