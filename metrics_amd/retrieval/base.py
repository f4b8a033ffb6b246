"""RetrievalMetric base — per-query grouping then averaging.

Parity: torchmetrics ``retrieval/base.py``: list states ``indexes/preds/
target`` (cat-reduced), compute sorts by index and applies the per-query
``_metric`` with ``empty_target_action`` in {'neg','pos','skip','error'}.

MI355X note: grouping is a single device sort + bincount split; the per-query
scoring loop runs on the already-sorted slices.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, List, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.utilities.checks import _check_retrieval_inputs
from metrics_amd.utilities.data import _flexible_bincount, dim_zero_cat


def _retrieval_aggregate(values: Tensor, aggregation="mean", dim: Optional[int] = None) -> Tensor:
    """Aggregate per-query retrieval scores: mean/median/min/max or a callable.

    Parity: reference retrieval/base.py:26 (_retrieval_aggregate).
    """
    if aggregation == "mean":
        return values.mean() if dim is None else values.mean(dim=dim)
    if aggregation == "median":
        return values.median() if dim is None else values.median(dim=dim).values
    if aggregation == "min":
        return values.min() if dim is None else values.min(dim=dim).values
    if aggregation == "max":
        return values.max() if dim is None else values.max(dim=dim).values
    return aggregation(values, dim=dim)


class _Grouped:
    """All-queries grouping for the batched retrieval path.

    Elements are sorted by (query index asc, pred desc). Fields:
    ``preds/target`` sorted that way; ``preds_by_index/target_by_index`` sorted
    by index only (loop fallback order); ``gid`` group id per element; ``ranks``
    0-based rank of each element within its query; ``counts`` per-query sizes;
    ``starts`` per-query offsets; ``npos/nneg/tsum`` per-query target stats.
    """

    __slots__ = ("preds", "target", "preds_by_index", "target_by_index",
                 "gid", "ranks", "counts", "starts", "npos", "nneg", "tsum", "G")

    def seg_sum(self, vals: Tensor) -> Tensor:
        out = torch.zeros(self.G, device=vals.device, dtype=vals.dtype)
        out.index_add_(0, self.gid, vals)
        return out


def _group_by_query(indexes: Tensor, preds: Tensor, target: Tensor) -> _Grouped:
    g = _Grouped()
    if indexes.is_cuda and indexes.numel() and int(indexes.max()) < 2**31:
        # ONE composite-key rocPRIM radix pass for the whole lexsort
        # (csrc/clf_curve.hip ma_retrieval_sort); one more 32-bit sort gives
        # the stable by-index order
        from metrics_amd.ops import _hip

        order, by_index = _hip.retrieval_sort(indexes, preds)
    else:
        # lexsort: stable pred-desc, then stable index-asc => index asc, pred desc
        order1 = torch.argsort(preds, descending=True, stable=True)
        idx_sorted_key = indexes[order1]
        order2 = torch.argsort(idx_sorted_key, stable=True)
        order = order1[order2]
        by_index = torch.argsort(indexes, stable=True)
    g.preds = preds[order]
    g.target = target[order]
    gsorted = indexes[order]
    # loop-fallback order (index asc, original order otherwise — like torch.sort)
    g.preds_by_index = preds[by_index]
    g.target_by_index = target[by_index]

    g.counts = _flexible_bincount(gsorted)
    g.G = int(g.counts.numel())
    dev = preds.device
    g.starts = torch.zeros(g.G, device=dev, dtype=torch.long)
    if g.G > 1:
        g.starts[1:] = torch.cumsum(g.counts, 0)[:-1]
    boundary = torch.zeros_like(gsorted, dtype=torch.long)
    if gsorted.numel() > 1:
        boundary[1:] = (gsorted[1:] != gsorted[:-1]).long()
    g.gid = torch.cumsum(boundary, 0)
    g.ranks = torch.arange(gsorted.numel(), device=dev) - g.starts[g.gid]
    tf = g.target.float()
    g.tsum = g.seg_sum(tf)
    g.npos = g.seg_sum((tf > 0).float())
    g.nneg = g.seg_sum((tf == 0).float())
    return g


class RetrievalMetric(Metric, ABC):
    """Base class for retrieval metrics over (indexes, preds, target) triplets."""

    is_differentiable: bool = False
    higher_is_better: bool = True
    full_state_update: bool = False

    indexes: List[Tensor]
    preds: List[Tensor]
    target: List[Tensor]

    def __init__(
        self,
        empty_target_action: str = "neg",
        ignore_index: Optional[int] = None,
        aggregation="mean",
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.allow_non_binary_target = False
        if not (aggregation in ("mean", "median", "min", "max") or callable(aggregation)):
            raise ValueError(
                "Argument `aggregation` must be one of `mean`, `median`, `min`, `max` or a custom callable function"
                f"which takes tensor of values, but got {aggregation}."
            )
        self.aggregation = aggregation

        empty_target_action_options = ("error", "skip", "neg", "pos")
        if empty_target_action not in empty_target_action_options:
            raise ValueError(f"Argument `empty_target_action` received a wrong value `{empty_target_action}`.")
        self.empty_target_action = empty_target_action

        if ignore_index is not None and not isinstance(ignore_index, int):
            raise ValueError("Argument `ignore_index` must be an integer or None.")
        self.ignore_index = ignore_index

        self.add_state("indexes", default=[], dist_reduce_fx=None)
        self.add_state("preds", default=[], dist_reduce_fx=None)
        self.add_state("target", default=[], dist_reduce_fx=None)

    def update(self, preds: Tensor, target: Tensor, indexes: Tensor) -> None:
        """Accumulate (indexes, preds, target)."""
        if indexes is None:
            raise ValueError("Argument `indexes` cannot be None")
        indexes, preds, target = _check_retrieval_inputs(
            indexes, preds, target, allow_non_binary_target=self.allow_non_binary_target, ignore_index=self.ignore_index
        )
        self.indexes.append(indexes)
        self.preds.append(preds)
        self.target.append(target)

    # batched path: which per-query condition triggers empty_target_action
    _empty_on_negatives: bool = False

    def compute(self) -> Tensor:
        """Group by query index and average the per-query metric.

        Fast path: subclasses implementing ``_batched_scores`` compute ALL
        queries with a handful of segmented tensor ops (sort + index_add) —
        no per-query python loop, so 100k queries cost the same handful of
        kernel launches as 10.
        """
        indexes = dim_zero_cat(self.indexes)
        preds = dim_zero_cat(self.preds)
        target = dim_zero_cat(self.target)

        grouped = _group_by_query(indexes, preds, target)
        scores = self._batched_scores(grouped)
        if scores is not None:
            empty = (grouped.nneg == 0) if self._empty_on_negatives else (grouped.tsum == 0)
            if bool(empty.any()):
                kind = "negative" if self._empty_on_negatives else "positive"
                if self.empty_target_action == "error":
                    raise ValueError(f"`compute` method was provided with a query with no {kind} target.")
                if self.empty_target_action == "skip":
                    scores = scores[~empty]
                else:
                    fill = 1.0 if self.empty_target_action == "pos" else 0.0
                    scores = torch.where(empty, torch.tensor(fill, device=scores.device), scores)
            if not scores.numel():
                return torch.tensor(0.0).to(preds)
            return _retrieval_aggregate(scores, self.aggregation).to(preds.dtype)

        # generic fallback: per-query loop over sorted slices
        split_sizes = grouped.counts.detach().cpu().tolist()
        res = []
        for mini_preds, mini_target in zip(
            torch.split(grouped.preds_by_index, split_sizes, dim=0),
            torch.split(grouped.target_by_index, split_sizes, dim=0),
        ):
            empty_q = ((1 - mini_target).sum() == 0) if self._empty_on_negatives else (not mini_target.sum())
            if empty_q:
                kind = "negative" if self._empty_on_negatives else "positive"
                if self.empty_target_action == "error":
                    raise ValueError(f"`compute` method was provided with a query with no {kind} target.")
                if self.empty_target_action == "pos":
                    res.append(torch.tensor(1.0, device=preds.device))
                elif self.empty_target_action == "neg":
                    res.append(torch.tensor(0.0, device=preds.device))
            else:
                res.append(self._metric(mini_preds, mini_target))

        if not res:
            return torch.tensor(0.0).to(preds)
        return _retrieval_aggregate(torch.stack([x.to(preds) for x in res]), self.aggregation)

    def _batched_scores(self, grouped: "_Grouped") -> Optional[Tensor]:
        """Vectorized per-query scores (G,), or None to use the per-query loop."""
        return None

    @abstractmethod
    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        """Score a single query's (preds, target)."""

    def plot(self, val: Optional[Any] = None, ax: Optional[Any] = None):
        return self._plot(val, ax)
