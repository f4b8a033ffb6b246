"""Multilabel ranking metrics. Parity: torchmetrics ``functional/classification/ranking.py``."""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.functional.classification.confusion_matrix import (
    _multilabel_confusion_matrix_arg_validation,
    _multilabel_confusion_matrix_format,
)
from metrics_amd.functional.classification.stat_scores import _multilabel_stat_scores_tensor_validation


def _rank_data(x: Tensor) -> Tensor:
    """Rank of each element (1 = smallest); ties get the same (min-style) rank via unique inverse."""
    _, inverse, counts = torch.unique(x, sorted=True, return_inverse=True, return_counts=True)
    ranks = counts.cumsum(dim=0)
    return ranks[inverse]


def _multilabel_ranking_format(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    ignore_index: Optional[int] = None,
) -> Tuple[Tensor, Tensor]:
    preds, target = _multilabel_confusion_matrix_format(
        preds, target, num_labels, threshold=0.0, ignore_index=ignore_index, should_threshold=False
    )
    return preds, target


def _multilabel_coverage_error_update(preds: Tensor, target: Tensor) -> Tuple[Tensor, int]:
    """How far along the score-sorted label list one must go to cover all true labels."""
    offset = torch.zeros_like(preds)
    offset[target == 0] = preds.min().abs() + 10  # Any number >1 works
    preds_mod = preds + offset
    preds_min = preds_mod.min(dim=1)[0]
    coverage = (preds >= preds_min[:, None]).sum(dim=1).to(torch.float32)
    return coverage.sum(), coverage.numel()


def multilabel_coverage_error(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Multilabel coverage error."""
    if validate_args:
        _multilabel_confusion_matrix_arg_validation(num_labels, threshold=0.0, ignore_index=ignore_index)
        _multilabel_stat_scores_tensor_validation(preds, target, num_labels, "global", ignore_index)
    preds, target = _multilabel_ranking_format(preds, target, num_labels, ignore_index)
    coverage, total = _multilabel_coverage_error_update(preds, target)
    return coverage / total


def _multilabel_ranking_average_precision_update(preds: Tensor, target: Tensor) -> Tuple[Tensor, int]:
    """Label ranking average precision (samplewise mean of per-label precision at each true label)."""
    neg_preds = -preds
    score = torch.tensor(0.0, device=neg_preds.device)
    n_preds, n_labels = neg_preds.shape
    for i in range(n_preds):
        relevant = target[i] == 1
        ranking = _rank_data(neg_preds[i][relevant]).float()
        if len(ranking) > 0 and len(ranking) < n_labels:
            rank = _rank_data(neg_preds[i])[relevant].float()
            score_idx = (ranking / rank).mean()
        else:
            score_idx = torch.ones_like(score)
        score += score_idx
    return score, n_preds


def multilabel_ranking_average_precision(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Multilabel ranking average precision."""
    if validate_args:
        _multilabel_confusion_matrix_arg_validation(num_labels, threshold=0.0, ignore_index=ignore_index)
        _multilabel_stat_scores_tensor_validation(preds, target, num_labels, "global", ignore_index)
    preds, target = _multilabel_ranking_format(preds, target, num_labels, ignore_index)
    score, total = _multilabel_ranking_average_precision_update(preds, target)
    return score / total


def _multilabel_ranking_loss_update(preds: Tensor, target: Tensor) -> Tuple[Tensor, int]:
    """Average number of wrongly-ordered (true, false) label pairs, weighted."""
    n_preds, n_labels = preds.shape
    relevant = target == 1
    n_relevant = relevant.sum(dim=1)

    # rows that are all-negative or all-positive carry no ranking signal; drop them
    mask = (n_relevant > 0) & (n_relevant < n_labels)
    preds = preds[mask]
    relevant = relevant[mask]
    n_relevant = n_relevant[mask]

    # Nothing is relevant
    if len(preds) == 0:
        return torch.tensor(0.0, device=preds.device), 1

    inverse = preds.argsort(dim=1).argsort(dim=1)
    per_label_loss = ((n_labels - inverse) * relevant).to(torch.float32)
    correction = 0.5 * n_relevant * (n_relevant + 1)
    denom = n_relevant * (n_labels - n_relevant)
    loss = (per_label_loss.sum(dim=1) - correction) / denom
    return loss.sum(), n_preds


def multilabel_ranking_loss(
    preds: Tensor,
    target: Tensor,
    num_labels: int,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Multilabel ranking loss."""
    if validate_args:
        _multilabel_confusion_matrix_arg_validation(num_labels, threshold=0.0, ignore_index=ignore_index)
        _multilabel_stat_scores_tensor_validation(preds, target, num_labels, "global", ignore_index)
    preds, target = _multilabel_ranking_format(preds, target, num_labels, ignore_index)
    loss, total = _multilabel_ranking_loss_update(preds, target)
    return loss / total
