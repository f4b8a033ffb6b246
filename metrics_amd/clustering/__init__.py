"""Modular clustering metrics. Parity: torchmetrics ``clustering/*``."""
from __future__ import annotations

from typing import Any, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.data import dim_zero_cat
from metrics_amd.functional.clustering.metrics import (
    adjusted_mutual_info_score,
    adjusted_rand_score,
    calinski_harabasz_score,
    completeness_score,
    davies_bouldin_score,
    dunn_index,
    fowlkes_mallows_index,
    homogeneity_score,
    mutual_info_score,
    normalized_mutual_info_score,
    rand_score,
    v_measure_score,
)


class _LabelPairMetric(Metric):
    """Base: accumulate (preds, target) label tensors, score at compute."""

    is_differentiable = True
    higher_is_better = True
    full_state_update: bool = True
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    preds: List[Tensor]
    target: List[Tensor]

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.add_state("preds", default=[], dist_reduce_fx="cat")
        self.add_state("target", default=[], dist_reduce_fx="cat")

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Append predicted and true cluster labels."""
        self.preds.append(preds)
        self.target.append(target)

    def compute(self) -> Tensor:
        return self._score(dim_zero_cat(self.preds), dim_zero_cat(self.target))

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class _EmbeddingMetric(Metric):
    """Base: accumulate (data, labels), score at compute."""

    is_differentiable = True
    higher_is_better = True
    full_state_update: bool = True
    plot_lower_bound: float = 0.0

    data: List[Tensor]
    labels: List[Tensor]

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.add_state("data", default=[], dist_reduce_fx="cat")
        self.add_state("labels", default=[], dist_reduce_fx="cat")

    def update(self, data: Tensor, labels: Tensor) -> None:
        """Append embeddings and their cluster labels."""
        self.data.append(data)
        self.labels.append(labels)

    def compute(self) -> Tensor:
        return self._score(dim_zero_cat(self.data), dim_zero_cat(self.labels))

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class MutualInfoScore(_LabelPairMetric):
    """Mutual information (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return mutual_info_score(a, b)


class RandScore(_LabelPairMetric):
    """Rand index (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return rand_score(a, b)


class AdjustedRandScore(_LabelPairMetric):
    """Adjusted Rand index (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return adjusted_rand_score(a, b)


class FowlkesMallowsIndex(_LabelPairMetric):
    """Fowlkes-Mallows index (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return fowlkes_mallows_index(a, b)


class HomogeneityScore(_LabelPairMetric):
    """Homogeneity (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return homogeneity_score(a, b)


class CompletenessScore(_LabelPairMetric):
    """Completeness (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return completeness_score(a, b)


class CalinskiHarabaszScore(_EmbeddingMetric):
    """Calinski-Harabasz score (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return calinski_harabasz_score(a, b)


class DaviesBouldinScore(_EmbeddingMetric):
    """Davies-Bouldin score (stateful)."""

    @staticmethod
    def _score(a: Tensor, b: Tensor) -> Tensor:
        return davies_bouldin_score(a, b)


class AdjustedMutualInfoScore(_LabelPairMetric):
    """Adjusted mutual information (stateful)."""

    def __init__(self, average_method: str = "arithmetic", **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if average_method not in ("min", "max", "arithmetic", "geometric"):
            raise ValueError(f"Expected argument `average_method` to be one of min/max/arithmetic/geometric but got {average_method}")
        self.average_method = average_method

    def _score(self, a: Tensor, b: Tensor) -> Tensor:
        return adjusted_mutual_info_score(a, b, self.average_method)


class NormalizedMutualInfoScore(_LabelPairMetric):
    """Normalized mutual information (stateful)."""

    def __init__(self, average_method: str = "arithmetic", **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if average_method not in ("min", "max", "arithmetic", "geometric"):
            raise ValueError(f"Expected argument `average_method` to be one of min/max/arithmetic/geometric but got {average_method}")
        self.average_method = average_method

    def _score(self, a: Tensor, b: Tensor) -> Tensor:
        return normalized_mutual_info_score(a, b, self.average_method)


class VMeasureScore(_LabelPairMetric):
    """V-measure (stateful)."""

    def __init__(self, beta: float = 1.0, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not (isinstance(beta, (int, float)) and beta > 0):
            raise ValueError(f"Expected argument `beta` to be a positive float but got {beta}")
        self.beta = float(beta)

    def _score(self, a: Tensor, b: Tensor) -> Tensor:
        return v_measure_score(a, b, self.beta)


class DunnIndex(_EmbeddingMetric):
    """Dunn index (stateful)."""

    def __init__(self, p: float = 2, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.p = p

    def _score(self, a: Tensor, b: Tensor) -> Tensor:
        return dunn_index(a, b, self.p)


__all__ = [
    "AdjustedMutualInfoScore",
    "AdjustedRandScore",
    "CalinskiHarabaszScore",
    "CompletenessScore",
    "DaviesBouldinScore",
    "DunnIndex",
    "FowlkesMallowsIndex",
    "HomogeneityScore",
    "MutualInfoScore",
    "NormalizedMutualInfoScore",
    "RandScore",
    "VMeasureScore",
]
