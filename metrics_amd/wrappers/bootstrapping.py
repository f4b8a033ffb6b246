"""BootStrapper. Parity: torchmetrics ``wrappers/bootstrapping.py``."""
from __future__ import annotations

from copy import deepcopy
from typing import Any, Dict, Optional, Sequence, Union

import torch
from torch import Tensor
from torch.nn import ModuleList

from metrics_amd.metric import Metric
from metrics_amd.wrappers.abstract import WrapperMetric


def _bootstrap_sampler(size: int, sampling_strategy: str = "poisson") -> Tensor:
    """Sample indices (with replacement) for one bootstrap replicate."""
    if sampling_strategy == "poisson":
        p = torch.distributions.Poisson(1)
        n = p.sample((size,))
        return torch.arange(size).repeat_interleave(n.long(), dim=0)
    if sampling_strategy == "multinomial":
        return torch.multinomial(torch.ones(size), num_samples=size, replacement=True)
    raise ValueError("Unknown sampling strategy")


class BootStrapper(WrapperMetric):
    """Maintain ``num_bootstraps`` resampled copies of a metric for uncertainty estimates."""

    full_state_update: Optional[bool] = True

    def __init__(
        self,
        base_metric: Metric,
        num_bootstraps: int = 10,
        mean: bool = True,
        std: bool = True,
        quantile: Optional[Union[float, Tensor]] = None,
        raw: bool = False,
        sampling_strategy: str = "poisson",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if not isinstance(base_metric, Metric):
            raise ValueError(
                f"Expected base metric to be an instance of metrics_amd.Metric but received {base_metric}"
            )
        self.metrics = ModuleList([deepcopy(base_metric) for _ in range(num_bootstraps)])
        self.num_bootstraps = num_bootstraps

        self.mean = mean
        self.std = std
        self.quantile = quantile
        self.raw = raw

        allowed_sampling = ("poisson", "multinomial")
        if sampling_strategy not in allowed_sampling:
            raise ValueError(
                f"Expected argument ``sampling_strategy`` to be one of {allowed_sampling} but received {sampling_strategy}"
            )
        self.sampling_strategy = sampling_strategy

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Update each bootstrap replicate on a resampled batch."""
        args_sizes = [a.shape[0] for a in args if isinstance(a, Tensor)]
        kwargs_sizes = [v.shape[0] for v in kwargs.values() if isinstance(v, Tensor)]
        if args_sizes:
            size = args_sizes[0]
        elif kwargs_sizes:
            size = kwargs_sizes[0]
        else:
            raise ValueError("None of the input contained tensors, so no sampling could be done")

        for idx in range(self.num_bootstraps):
            sample_idx = _bootstrap_sampler(size, sampling_strategy=self.sampling_strategy)
            if sample_idx.numel() == 0:
                continue
            new_args = [a[sample_idx] if isinstance(a, Tensor) else a for a in args]
            new_kwargs = {k: v[sample_idx] if isinstance(v, Tensor) else v for k, v in kwargs.items()}
            self.metrics[idx].update(*new_args, **new_kwargs)

    def compute(self) -> Dict[str, Tensor]:
        """Mean/std/quantile/raw over the bootstrap replicates."""
        computed_vals = torch.stack([m.compute() for m in self.metrics], dim=0)
        output_dict = {}
        if self.mean:
            output_dict["mean"] = computed_vals.mean(dim=0)
        if self.std:
            output_dict["std"] = computed_vals.std(dim=0)
        if self.quantile is not None:
            output_dict["quantile"] = torch.quantile(computed_vals, self.quantile)
        if self.raw:
            output_dict["raw"] = computed_vals
        return output_dict

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        self.update(*args, **kwargs)
        self._forward_cache = self.compute()
        return self._forward_cache

    def reset(self) -> None:
        for m in self.metrics:
            m.reset()
        super().reset()

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)
