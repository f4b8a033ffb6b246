"""ROUGE. Parity: torchmetrics ``functional/text/rouge.py`` (rouge1/2/.../L/Lsum).

Normalization mirrors the rouge-score package (non-alphanumeric strip,
lowercase); Porter stemming is applied when nltk is available.
"""
from __future__ import annotations

import re
from collections import Counter
from typing import Dict, List, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor, tensor

from metrics_amd.utilities.imports import _NLTK_AVAILABLE

ALLOWED_ROUGE_KEYS = {
    "rouge1": 1,
    "rouge2": 2,
    "rouge3": 3,
    "rouge4": 4,
    "rouge5": 5,
    "rouge6": 6,
    "rouge7": 7,
    "rouge8": 8,
    "rouge9": 9,
    "rougeL": "L",
    "rougeLsum": "Lsum",
}
ALLOWED_ACCUMULATE_VALUES = ("avg", "best")


def _normalize_text(text: str, stemmer=None) -> List[str]:
    text = re.sub(r"[^a-z0-9]+", " ", text.lower())
    tokens = text.split()
    if stemmer is not None:
        tokens = [stemmer.stem(t) if len(t) > 3 else t for t in tokens]
    return tokens


def _split_sentences(text: str) -> List[str]:
    return [s for s in re.split(r"[.!?]\s*|\n", text) if s.strip()]


def _ngram_counts(tokens: Sequence[str], n: int) -> Counter:
    return Counter(tuple(tokens[i : i + n]) for i in range(len(tokens) - n + 1))


def _fmeasure(matches: int, pred_total: int, target_total: int) -> Dict[str, Tensor]:
    precision = matches / pred_total if pred_total > 0 else 0.0
    recall = matches / target_total if target_total > 0 else 0.0
    if precision + recall > 0:
        fmeasure = 2 * precision * recall / (precision + recall)
    else:
        fmeasure = 0.0
    return {"precision": tensor(precision), "recall": tensor(recall), "fmeasure": tensor(fmeasure)}


def _lcs_len(a: Sequence[str], b: Sequence[str]) -> int:
    n, m = len(a), len(b)
    if n == 0 or m == 0:
        return 0
    prev = [0] * (m + 1)
    for i in range(1, n + 1):
        cur = [0] * (m + 1)
        ai = a[i - 1]
        for j in range(1, m + 1):
            if ai == b[j - 1]:
                cur[j] = prev[j - 1] + 1
            else:
                cur[j] = max(prev[j], cur[j - 1])
        prev = cur
    return prev[m]


def _union_lcs(pred_sentences: List[List[str]], target_sentence: List[str]) -> set:
    """Union of LCS token positions for rougeLsum (greedy union approximation)."""
    hits = set()
    for ps in pred_sentences:
        # mark matched target indices of the lcs
        n, m = len(ps), len(target_sentence)
        dp = [[0] * (m + 1) for _ in range(n + 1)]
        for i in range(1, n + 1):
            for j in range(1, m + 1):
                if ps[i - 1] == target_sentence[j - 1]:
                    dp[i][j] = dp[i - 1][j - 1] + 1
                else:
                    dp[i][j] = max(dp[i - 1][j], dp[i][j - 1])
        i, j = n, m
        while i > 0 and j > 0:
            if ps[i - 1] == target_sentence[j - 1] and dp[i][j] == dp[i - 1][j - 1] + 1:
                hits.add(j - 1)
                i -= 1
                j -= 1
            elif dp[i - 1][j] >= dp[i][j - 1]:
                i -= 1
            else:
                j -= 1
    return hits


def _rouge_score_one(pred: str, target: str, rouge_keys, stemmer) -> Dict[str, Dict[str, Tensor]]:
    out = {}
    p_tokens = _normalize_text(pred, stemmer)
    t_tokens = _normalize_text(target, stemmer)
    for key, n in rouge_keys.items():
        if isinstance(n, int):
            p_ng = _ngram_counts(p_tokens, n)
            t_ng = _ngram_counts(t_tokens, n)
            matches = sum((p_ng & t_ng).values())
            out[key] = _fmeasure(matches, sum(p_ng.values()), sum(t_ng.values()))
        elif n == "L":
            lcs = _lcs_len(p_tokens, t_tokens)
            out[key] = _fmeasure(lcs, len(p_tokens), len(t_tokens))
        else:  # Lsum
            p_sents = [_normalize_text(s, stemmer) for s in _split_sentences(pred)]
            t_sents = [_normalize_text(s, stemmer) for s in _split_sentences(target)]
            matches = sum(len(_union_lcs(p_sents, ts)) for ts in t_sents)
            out[key] = _fmeasure(matches, sum(len(s) for s in p_sents), sum(len(s) for s in t_sents))
    return out


def rouge_score(
    preds: Union[str, Sequence[str]],
    target: Union[str, Sequence[str], Sequence[Sequence[str]]],
    accumulate: str = "best",
    use_stemmer: bool = False,
    normalizer=None,
    tokenizer=None,
    rouge_keys: Union[str, Tuple[str, ...]] = ("rouge1", "rouge2", "rougeL", "rougeLsum"),
) -> Dict[str, Tensor]:
    """ROUGE scores (precision/recall/fmeasure per key, averaged over corpus)."""
    if use_stemmer and not _NLTK_AVAILABLE:
        raise ModuleNotFoundError("Stemmer requires `nltk` which is not installed.")
    stemmer = None
    if use_stemmer:
        import nltk

        stemmer = nltk.stem.porter.PorterStemmer()

    if isinstance(rouge_keys, str):
        rouge_keys = (rouge_keys,)
    for key in rouge_keys:
        if key not in ALLOWED_ROUGE_KEYS:
            raise ValueError(f"Got unknown rouge key {key}. Expected to be one of {list(ALLOWED_ROUGE_KEYS)}")
    keys = {k: ALLOWED_ROUGE_KEYS[k] for k in rouge_keys}
    if accumulate not in ALLOWED_ACCUMULATE_VALUES:
        raise ValueError(f"Got unknown accumulate value {accumulate}. Expected to be one of {ALLOWED_ACCUMULATE_VALUES}")

    preds_ = [preds] if isinstance(preds, str) else list(preds)
    if isinstance(target, str):
        target_: List[List[str]] = [[target]]
    elif target and isinstance(target[0], str):
        target_ = [[t] for t in target]
    else:
        target_ = [list(t) for t in target]

    agg: Dict[str, Dict[str, List[Tensor]]] = {k: {"precision": [], "recall": [], "fmeasure": []} for k in keys}
    for p, refs in zip(preds_, target_):
        per_ref = [_rouge_score_one(p, r, keys, stemmer) for r in refs]
        for k in keys:
            if accumulate == "best":
                best = max(per_ref, key=lambda d: d[k]["fmeasure"])
                chosen = best[k]
            else:  # avg
                chosen = {
                    m: torch.stack([d[k][m] for d in per_ref]).mean() for m in ("precision", "recall", "fmeasure")
                }
            for m in ("precision", "recall", "fmeasure"):
                agg[k][m].append(chosen[m])

    out: Dict[str, Tensor] = {}
    for k in keys:
        for m in ("precision", "recall", "fmeasure"):
            out[f"{k}_{m}"] = torch.stack(agg[k][m]).mean()
    return out
