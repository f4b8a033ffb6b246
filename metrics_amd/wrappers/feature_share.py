"""FeatureShare. Parity: torchmetrics ``wrappers/feature_share.py``.

Lets several network-backed metrics (FID/KID/IS...) share ONE feature
extractor forward per batch via an lru-cached network wrapper.
"""
from __future__ import annotations

from functools import lru_cache
from typing import Any, Dict, List, Optional, Sequence, Union

import torch
from torch import Tensor
from torch.nn import Module

from metrics_amd.collections import MetricCollection
from metrics_amd.metric import Metric


class NetworkCache(Module):
    """Cache the forward output of a network by input-tensor identity."""

    def __init__(self, network: Module, max_size: int = 100) -> None:
        super().__init__()
        self.max_size = max_size
        self.network = network
        self.network.forward = lru_cache(maxsize=self.max_size)(network.forward)

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        return self.network(*args, **kwargs)


def _get_network_attr_name(metric: Metric) -> str:
    for name in ("inception", "net", "model", "feature_network"):
        if hasattr(metric, name) and isinstance(getattr(metric, name), Module):
            return name
    # fallback: declared attribute
    if getattr(metric, "feature_network", None) and isinstance(getattr(metric, metric.feature_network, None), Module):
        return metric.feature_network
    raise AttributeError(f"Could not find a network attribute on metric {metric.__class__.__name__}")


class FeatureShare(MetricCollection):
    """MetricCollection whose members share one cached feature-extractor."""

    def __init__(
        self,
        metrics: Union[Metric, Sequence[Metric], Dict[str, Metric]],
        max_cache_size: Optional[int] = None,
    ) -> None:
        super().__init__(metrics=metrics, compute_groups=False)

        if max_cache_size is None:
            max_cache_size = len(self)
        if not isinstance(max_cache_size, int):
            raise TypeError(f"max_cache_size should be an integer, but got {max_cache_size}")

        try:
            first = next(iter(self.values(copy_state=False)))
            attr = _get_network_attr_name(first)
            network_to_share = getattr(first, attr)
        except AttributeError as err:
            raise AttributeError(
                "Tried to extract the network to share from the first metric, but it did not have a network attribute."
            ) from err
        shared_net = NetworkCache(network_to_share, max_size=max_cache_size)

        for metric in self.values(copy_state=False):
            attr = _get_network_attr_name(metric)
            setattr(metric, attr, shared_net)
