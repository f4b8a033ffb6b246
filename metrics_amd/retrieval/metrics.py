"""Modular retrieval metrics. Parity: torchmetrics ``retrieval/*``."""
from __future__ import annotations

from typing import Any, Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.retrieval.base import RetrievalMetric
from metrics_amd.utilities.data import _flexible_bincount, dim_zero_cat
from metrics_amd.functional.retrieval.metrics import (
    retrieval_auroc,
    retrieval_average_precision,
    retrieval_fall_out,
    retrieval_hit_rate,
    retrieval_normalized_dcg,
    retrieval_precision,
    retrieval_precision_recall_curve,
    retrieval_r_precision,
    retrieval_recall,
    retrieval_reciprocal_rank,
)


class _TopKRetrievalMetric(RetrievalMetric):
    """Shared top_k handling."""

    def __init__(self, empty_target_action: str = "neg", ignore_index: Optional[int] = None,
                 top_k: Optional[int] = None, aggregation="mean", **kwargs: Any) -> None:
        super().__init__(empty_target_action=empty_target_action, ignore_index=ignore_index,
                         aggregation=aggregation, **kwargs)
        if top_k is not None and not (isinstance(top_k, int) and top_k > 0):
            raise ValueError("`top_k` has to be a positive integer or None")
        self.top_k = top_k

    def _k_eff(self, g) -> Tensor:
        """Per-query truncation length: min(top_k, query size)."""
        return g.counts.clamp(max=self.top_k) if self.top_k is not None else g.counts

    def _in_k(self, g) -> Tensor:
        """Elementwise mask: element is within its query's top-k by pred."""
        return g.ranks < self._k_eff(g)[g.gid]


class RetrievalMAP(_TopKRetrievalMetric):
    """Mean average precision."""

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_average_precision(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        tf = (g.target.float() > 0).float()
        in_k = self._in_k(g).float()
        cum = torch.cumsum(tf, 0)
        base = (cum - tf)[g.starts]  # exclusive prefix at group start
        cum_in_group = cum - base[g.gid]  # inclusive positive count at each rank
        prec_at_pos = cum_in_group / (g.ranks + 1).float()
        hits_in_k = g.seg_sum(tf * in_k)
        ap_num = g.seg_sum(prec_at_pos * tf * in_k)
        return torch.where(hits_in_k > 0, ap_num / hits_in_k.clamp(min=1), torch.zeros_like(ap_num))


class RetrievalMRR(_TopKRetrievalMetric):
    """Mean reciprocal rank."""

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_reciprocal_rank(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        pos = g.target.float() > 0
        big = torch.iinfo(torch.long).max
        r0 = torch.full((g.G,), big, device=g.ranks.device, dtype=torch.long)
        r0.scatter_reduce_(0, g.gid[pos], g.ranks[pos], reduce="amin")
        found = r0 < self._k_eff(g)
        return torch.where(found, 1.0 / (r0.clamp(max=big - 1) + 1).float(), torch.zeros(g.G, device=g.ranks.device))


class RetrievalPrecision(_TopKRetrievalMetric):
    """Precision@k."""

    def __init__(self, empty_target_action: str = "neg", ignore_index: Optional[int] = None,
                 top_k: Optional[int] = None, adaptive_k: bool = False, aggregation="mean", **kwargs: Any) -> None:
        super().__init__(empty_target_action, ignore_index, top_k, aggregation=aggregation, **kwargs)
        if not isinstance(adaptive_k, bool):
            raise ValueError("`adaptive_k` has to be a boolean")
        self.adaptive_k = adaptive_k

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_precision(preds, target, top_k=self.top_k, adaptive_k=self.adaptive_k)

    def _batched_scores(self, g):
        # divisor: top_k (even past the query size) unless adaptive/None -> query size
        if self.top_k is None:
            div = g.counts.float()
        elif self.adaptive_k:
            div = g.counts.clamp(max=self.top_k).float()
        else:
            div = torch.full((g.G,), float(self.top_k), device=g.counts.device)
        hits = g.seg_sum((g.target.float() > 0).float() * self._in_k(g).float())
        return hits / div


class RetrievalRecall(_TopKRetrievalMetric):
    """Recall@k."""

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_recall(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        hits = g.seg_sum((g.target.float() > 0).float() * self._in_k(g).float())
        return hits / g.npos.clamp(min=1)


class RetrievalHitRate(_TopKRetrievalMetric):
    """Hit rate@k."""

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_hit_rate(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        hits = g.seg_sum((g.target.float() > 0).float() * self._in_k(g).float())
        return (hits > 0).float()


class RetrievalFallOut(_TopKRetrievalMetric):
    """Fall-out@k (lower is better; empty_target_action applies to queries with no NEGATIVES)."""

    higher_is_better = False
    _empty_on_negatives = True

    def __init__(self, empty_target_action: str = "pos", ignore_index: Optional[int] = None,
                 top_k: Optional[int] = None, aggregation="mean", **kwargs: Any) -> None:
        # reference default differs from the family: a query with no negatives
        # is a PERFECT fall-out, so the fill default is "pos" (retrieval/fall_out.py)
        super().__init__(empty_target_action, ignore_index, top_k, aggregation=aggregation, **kwargs)

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_fall_out(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        neg_hits = g.seg_sum((g.target.float() == 0).float() * self._in_k(g).float())
        return neg_hits / g.nneg.clamp(min=1)


class RetrievalNormalizedDCG(_TopKRetrievalMetric):
    """Normalized DCG (graded relevance allowed)."""

    def __init__(self, empty_target_action: str = "neg", ignore_index: Optional[int] = None,
                 top_k: Optional[int] = None, aggregation="mean", **kwargs: Any) -> None:
        super().__init__(empty_target_action, ignore_index, top_k, aggregation=aggregation, **kwargs)
        self.allow_non_binary_target = True

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_normalized_dcg(preds, target, top_k=self.top_k)

    def _batched_scores(self, g):
        tf = g.target.float()
        in_k = self._in_k(g).float()
        disc = torch.log2(g.ranks.float() + 2.0)
        dcg = g.seg_sum(tf / disc * in_k)
        # ideal ordering: stable lexsort (gid asc, target desc) reusing g.ranks grid
        ord1 = torch.argsort(tf, descending=True, stable=True)
        ord2 = torch.argsort(g.gid[ord1], stable=True)
        t_ideal = tf[ord1][ord2]
        idcg = g.seg_sum(t_ideal / disc * in_k)
        return torch.where(idcg > 0, dcg / idcg.clamp(min=1e-38), torch.zeros_like(dcg))


class RetrievalRPrecision(RetrievalMetric):
    """R-precision."""

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_r_precision(preds, target)

    def _batched_scores(self, g):
        in_r = (g.ranks < g.npos[g.gid].long()).float()
        hits = g.seg_sum((g.target.float() > 0).float() * in_r)
        return hits / g.npos.clamp(min=1)


class RetrievalAUROC(_TopKRetrievalMetric):
    """Per-query AUROC, averaged."""

    def __init__(self, empty_target_action: str = "neg", ignore_index: Optional[int] = None,
                 top_k: Optional[int] = None, max_fpr: Optional[float] = None, aggregation="mean", **kwargs: Any) -> None:
        super().__init__(empty_target_action, ignore_index, top_k, aggregation=aggregation, **kwargs)
        self.max_fpr = max_fpr

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        return retrieval_auroc(preds, target, top_k=self.top_k, max_fpr=self.max_fpr)


class RetrievalPrecisionRecallCurve(RetrievalMetric):
    """Averaged precision/recall at k = 1..max_k."""

    def __init__(self, max_k: Optional[int] = None, adaptive_k: bool = False,
                 empty_target_action: str = "neg", ignore_index: Optional[int] = None,
                 aggregation="mean", **kwargs: Any) -> None:
        super().__init__(empty_target_action=empty_target_action, ignore_index=ignore_index,
                         aggregation=aggregation, **kwargs)
        if max_k is not None and not (isinstance(max_k, int) and max_k > 0):
            raise ValueError("`max_k` has to be a positive integer or None")
        self.max_k = max_k
        if not isinstance(adaptive_k, bool):
            raise ValueError("`adaptive_k` has to be a boolean")
        self.adaptive_k = adaptive_k

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        raise NotImplementedError

    def compute(self) -> Tuple[Tensor, Tensor, Tensor]:
        indexes = dim_zero_cat(self.indexes)
        preds = dim_zero_cat(self.preds)
        target = dim_zero_cat(self.target)

        indexes, indices = torch.sort(indexes)
        preds = preds[indices]
        target = target[indices]
        split_sizes = _flexible_bincount(indexes).detach().cpu().tolist()

        max_k = self.max_k or max(split_sizes)
        precisions, recalls = [], []
        for mini_preds, mini_target in zip(
            torch.split(preds, split_sizes, dim=0), torch.split(target, split_sizes, dim=0)
        ):
            if not mini_target.sum():
                if self.empty_target_action == "error":
                    raise ValueError("`compute` method was provided with a query with no positive target.")
                if self.empty_target_action == "skip":
                    continue
                fill = 1.0 if self.empty_target_action == "pos" else 0.0
                precisions.append(torch.full((max_k,), fill, device=preds.device))
                recalls.append(torch.full((max_k,), fill, device=preds.device))
            else:
                p, r, _ = retrieval_precision_recall_curve(mini_preds, mini_target, max_k, self.adaptive_k)
                precisions.append(p)
                recalls.append(r)
        top_k = torch.arange(1, max_k + 1, device=preds.device)
        if not precisions:
            return torch.zeros(max_k, device=preds.device), torch.zeros(max_k, device=preds.device), top_k
        from metrics_amd.retrieval.base import _retrieval_aggregate

        return (
            _retrieval_aggregate(torch.stack(precisions), self.aggregation, dim=0),
            _retrieval_aggregate(torch.stack(recalls), self.aggregation, dim=0),
            top_k,
        )


class RetrievalRecallAtFixedPrecision(RetrievalMetric):
    """Max recall@k such that precision@k >= min_precision; returns (recall, k)."""

    higher_is_better = True

    def __init__(self, min_precision: float = 0.0, max_k: Optional[int] = None, adaptive_k: bool = False,
                 empty_target_action: str = "neg", ignore_index: Optional[int] = None, **kwargs: Any) -> None:
        super().__init__(empty_target_action=empty_target_action, ignore_index=ignore_index, **kwargs)
        if not (isinstance(min_precision, float) and 0.0 <= min_precision <= 1.0):
            raise ValueError("`min_precision` has to be a positive float between 0 and 1")
        self.min_precision = min_precision
        self.max_k = max_k
        self.adaptive_k = adaptive_k

    def _metric(self, preds: Tensor, target: Tensor) -> Tensor:
        raise NotImplementedError

    def compute(self) -> Tuple[Tensor, Tensor]:
        curve = RetrievalPrecisionRecallCurve.compute(self)  # reuse grouping logic
        precisions, recalls, top_k = curve
        condition = precisions >= self.min_precision
        if condition.any():
            recalls_at = recalls[condition]
            best = recalls_at.max()
            k = top_k[condition][recalls_at.argmax()]
            return best, k
        return torch.tensor(0.0, device=precisions.device), top_k[-1]
