"""Per-query retrieval metrics (functional).

Parity: torchmetrics ``functional/retrieval/*`` — each function scores ONE
query's (preds, target); the modular layer groups by ``indexes`` and averages.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_retrieval_functional_inputs


def _topk(preds: Tensor, top_k: Optional[int]) -> int:
    k = preds.shape[-1] if top_k is None else top_k
    if not (isinstance(k, int) and k > 0):
        raise ValueError("`top_k` has to be a positive integer or None")
    return min(k, preds.shape[-1])


def retrieval_average_precision(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Average precision for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    target = target[preds.topk(k, dim=-1).indices]
    if not target.sum():
        return torch.tensor(0.0, device=preds.device)
    positions = torch.arange(1, len(target) + 1, device=target.device, dtype=torch.float32)[target > 0]
    return torch.div((torch.arange(len(positions), device=positions.device, dtype=torch.float32) + 1), positions).mean()


def retrieval_reciprocal_rank(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Mean reciprocal rank for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    target = target[preds.topk(k, dim=-1).indices]
    if not target.sum():
        return torch.tensor(0.0, device=preds.device)
    position = torch.nonzero(target).view(-1)
    return 1.0 / (position[0] + 1)


def retrieval_precision(preds: Tensor, target: Tensor, top_k: Optional[int] = None, adaptive_k: bool = False) -> Tensor:
    """Precision@k for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    if top_k is None or (adaptive_k and top_k > preds.shape[-1]):
        top_k = preds.shape[-1]
    if not (isinstance(top_k, int) and top_k > 0):
        raise ValueError("`top_k` has to be a positive integer or None")
    if not target.sum():
        return torch.tensor(0.0, device=preds.device)
    relevant = target[preds.topk(min(top_k, preds.shape[-1]), dim=-1).indices].sum().float()
    return relevant / top_k


def retrieval_recall(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Recall@k for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    if not target.sum():
        return torch.tensor(0.0, device=preds.device)
    relevant = target[preds.topk(k, dim=-1).indices].sum().float()
    return relevant / target.sum()


def retrieval_hit_rate(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Hit rate@k for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    relevant = target[preds.topk(k, dim=-1).indices].sum()
    return (relevant > 0).float()


def retrieval_fall_out(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Fall-out@k (non-relevant retrieved / total non-relevant) for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    target = 1 - target
    if not target.sum():
        return torch.tensor(0.0, device=preds.device)
    relevant = target[preds.topk(k, dim=-1).indices].sum().float()
    return relevant / target.sum()


def retrieval_r_precision(preds: Tensor, target: Tensor) -> Tensor:
    """R-precision for a single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    r = target.sum()
    if not r:
        return torch.tensor(0.0, device=preds.device)
    relevant = target[preds.topk(int(r.item()), dim=-1).indices].sum().float()
    return relevant / r


def _dcg(scores: Tensor) -> Tensor:
    denom = torch.log2(torch.arange(scores.shape[-1], device=scores.device) + 2.0)
    return (scores / denom).sum(dim=-1)


def retrieval_normalized_dcg(preds: Tensor, target: Tensor, top_k: Optional[int] = None) -> Tensor:
    """Normalized discounted cumulative gain for a single query (graded relevance ok)."""
    preds, target = _check_retrieval_functional_inputs(preds, target, allow_non_binary_target=True)
    k = _topk(preds, top_k)
    sorted_target = target[torch.argsort(preds, dim=-1, descending=True)][:k]
    ideal_target = torch.sort(target, descending=True)[0][:k]

    ideal_dcg = _dcg(ideal_target)
    target_dcg = _dcg(sorted_target)

    # filter undefined scores
    all_irrelevant = ideal_dcg == 0
    if all_irrelevant:
        return torch.tensor(0.0, device=preds.device)
    return target_dcg / ideal_dcg


def retrieval_auroc(preds: Tensor, target: Tensor, top_k: Optional[int] = None, max_fpr: Optional[float] = None) -> Tensor:
    """AUROC for a single query."""
    from metrics_amd.functional.classification.auroc import binary_auroc

    preds, target = _check_retrieval_functional_inputs(preds, target)
    k = _topk(preds, top_k)
    idx = preds.topk(k, dim=-1).indices
    preds_k, target_k = preds[idx], target[idx]
    if target_k.sum() == 0 or target_k.sum() == target_k.numel():
        return torch.tensor(0.0, device=preds.device)
    return binary_auroc(preds_k, target_k, max_fpr=max_fpr)


def retrieval_precision_recall_curve(
    preds: Tensor, target: Tensor, max_k: Optional[int] = None, adaptive_k: bool = False
) -> Tuple[Tensor, Tensor, Tensor]:
    """(precision@k, recall@k, k) for k = 1..max_k, single query."""
    preds, target = _check_retrieval_functional_inputs(preds, target)
    if max_k is None:
        max_k = preds.shape[-1]
    if not (isinstance(max_k, int) and max_k > 0):
        raise ValueError("`max_k` has to be a positive integer or None")
    if adaptive_k and max_k > preds.shape[-1]:
        max_k = preds.shape[-1]
    k = min(max_k, preds.shape[-1])
    topk = torch.arange(1, max_k + 1, device=preds.device)
    relevant = target[preds.topk(k, dim=-1).indices]
    cum_rel = torch.cumsum(relevant, dim=0).float()
    if k < max_k:
        cum_rel = torch.cat([cum_rel, cum_rel[-1].repeat(max_k - k)])
    precision = cum_rel / topk
    recall = cum_rel / target.sum() if target.sum() else torch.zeros_like(precision)
    return precision, recall, topk
