"""CER / WER / MER / WIL / WIP / EditDistance.

Parity: torchmetrics ``functional/text/{cer,wer,mer,wil,wip,edit}.py``.
"""
from __future__ import annotations

from typing import List, Tuple, Union

import torch
from torch import Tensor, tensor

from metrics_amd.functional.text.helper import (
    _edit_distance,
    _edit_distance_batch,
    _edit_distance_counts,
    _edit_distance_counts_batch,
)


def _norm_inputs(preds: Union[str, List[str]], target: Union[str, List[str]]) -> Tuple[List[str], List[str]]:
    if isinstance(preds, str):
        preds = [preds]
    if isinstance(target, str):
        target = [target]
    return list(preds), list(target)


def _cer_update(preds, target) -> Tuple[Tensor, Tensor]:
    preds, target = _norm_inputs(preds, target)
    errors = tensor(0, dtype=torch.float)
    total = tensor(0, dtype=torch.float)
    dists = _edit_distance_batch(list(zip(preds, target)))  # str pairs: utf-32 fast path
    for d, t in zip(dists, target):
        errors += d
        total += len(t)
    return errors, total


def char_error_rate(preds, target) -> Tensor:
    """Character error rate."""
    errors, total = _cer_update(preds, target)
    return errors / total


def _wer_update(preds, target) -> Tuple[Tensor, Tensor]:
    preds, target = _norm_inputs(preds, target)
    errors = tensor(0, dtype=torch.float)
    total = tensor(0, dtype=torch.float)
    pairs = [(p.split(), t.split()) for p, t in zip(preds, target)]
    dists = _edit_distance_batch(pairs)
    for d, (_, t_tokens) in zip(dists, pairs):
        errors += d
        total += len(t_tokens)
    return errors, total


def word_error_rate(preds, target) -> Tensor:
    """Word error rate."""
    errors, total = _wer_update(preds, target)
    return errors / total


def _mer_wil_wip_update(preds, target) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Return (Σ edit distance, Σ max(len_t, len_p), Σ len_t, Σ len_p).

    Reference formulation (functional/text/{mer,wip}.py): the denominator is
    the max sentence length, NOT backtrace hit counts — the two differ when a
    pair needs both insertions and deletions.
    """
    preds, target = _norm_inputs(preds, target)
    errors = tensor(0, dtype=torch.float)
    total = tensor(0, dtype=torch.float)
    target_total = tensor(0, dtype=torch.float)
    preds_total = tensor(0, dtype=torch.float)
    pairs = [(p.split(), t.split()) for p, t in zip(preds, target)]
    for (p_tokens, t_tokens), d in zip(pairs, _edit_distance_batch(pairs)):
        errors += d
        total += max(len(t_tokens), len(p_tokens))
        target_total += len(t_tokens)
        preds_total += len(p_tokens)
    return errors, total, target_total, preds_total


def match_error_rate(preds, target) -> Tensor:
    """Match error rate."""
    errors, total, _, _ = _mer_wil_wip_update(preds, target)
    return errors / total


def word_information_lost(preds, target) -> Tensor:
    """Word information lost = 1 - WIP."""
    return 1 - word_information_preserved(preds, target)


def word_information_preserved(preds, target) -> Tensor:
    """Word information preserved = ((total - errors)/len_t) * ((total - errors)/len_p)."""
    errors, total, target_total, preds_total = _mer_wil_wip_update(preds, target)
    kept = total - errors
    return kept / target_total * (kept / preds_total)


def edit_distance(preds, target, substitution_cost: int = 1, reduction: str = "mean") -> Tensor:
    """Raw (character-level) Levenshtein distance between string pairs."""
    preds, target = _norm_inputs(preds, target)
    if substitution_cost != 1:
        # generalized DP with substitution cost
        def dist(a: str, b: str) -> int:
            n, m = len(a), len(b)
            prev = list(range(m + 1))
            cur = [0] * (m + 1)
            for i in range(1, n + 1):
                cur[0] = i
                for j in range(1, m + 1):
                    cost = 0 if a[i - 1] == b[j - 1] else substitution_cost
                    cur[j] = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + cost)
                prev, cur = cur, prev
            return prev[m]

        vals = [dist(p, t) for p, t in zip(preds, target)]
    else:
        vals = _edit_distance_batch(list(zip(preds, target)))  # str pairs: utf-32 fast path
    res = torch.tensor(vals, dtype=torch.int32)
    if reduction == "mean":
        return res.float().mean()
    if reduction == "sum":
        return res.sum()
    if reduction is None or reduction == "none":
        return res
    raise ValueError(f"Expected argument `reduction` to be one of 'mean', 'sum', 'none' but got {reduction}")
