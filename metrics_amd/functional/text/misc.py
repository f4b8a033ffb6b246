"""SQuAD, Perplexity, ExtendedEditDistance.

Parity: torchmetrics ``functional/text/{squad,perplexity,eed}.py``.
"""
from __future__ import annotations

import re
import string
from collections import Counter
from typing import Dict, List, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor, tensor


# ------------------------------------------------------------------------ SQuAD
def _normalize_answer(s: str) -> str:
    """Lowercase, strip punctuation/articles, normalize whitespace."""
    s = s.lower()
    s = "".join(ch for ch in s if ch not in set(string.punctuation))
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


def _squad_f1(pred: str, truth: str) -> float:
    pred_tokens = _normalize_answer(pred).split()
    truth_tokens = _normalize_answer(truth).split()
    common = Counter(pred_tokens) & Counter(truth_tokens)
    num_same = sum(common.values())
    if num_same == 0:
        return 0.0
    precision = num_same / len(pred_tokens)
    recall = num_same / len(truth_tokens)
    return 2 * precision * recall / (precision + recall)


def _squad_em(pred: str, truth: str) -> float:
    return float(_normalize_answer(pred) == _normalize_answer(truth))


def squad(preds, target) -> Dict[str, Tensor]:
    """SQuAD v1 exact-match + F1.

    ``preds``: dict or list of dicts {"prediction_text", "id"};
    ``target``: dict or list of dicts {"answers": {"text": [...]}, "id"}.
    """
    if isinstance(preds, dict):
        preds = [preds]
    if isinstance(target, dict):
        target = [target]
    pred_by_id = {p["id"]: p["prediction_text"] for p in preds}
    f1_total, em_total, count = 0.0, 0.0, 0
    for t in target:
        tid = t["id"]
        if tid not in pred_by_id:
            continue
        answers = t["answers"]["text"]
        pred_text = pred_by_id[tid]
        f1_total += max(_squad_f1(pred_text, a) for a in answers)
        em_total += max(_squad_em(pred_text, a) for a in answers)
        count += 1
    return {
        "exact_match": tensor(100.0 * em_total / count if count else 0.0),
        "f1": tensor(100.0 * f1_total / count if count else 0.0),
    }


# -------------------------------------------------------------------- Perplexity
def _perplexity_update(preds: Tensor, target: Tensor, ignore_index: Optional[int] = None) -> Tuple[Tensor, Tensor]:
    """Accumulate total negative log likelihood + token count from logits."""
    if preds.ndim != 3:
        raise ValueError(f"Input tensor `preds` is expected to have 3 dimensions, [batch_size, seq_len, vocab_size], but got {preds.ndim}.")
    if target.ndim != 2:
        raise ValueError(f"Input tensor `target` is expected to have 2 dimensions, [batch_size, seq_len], but got {target.ndim}.")
    if preds.shape[:2] != target.shape:
        raise ValueError(
            "Input tensors `preds` and `target` are expected to have equaling first two dimensions,"
            f" [batch_size, seq_len], but got {preds.shape[:2]} and {target.shape}."
        )

    probs = torch.nn.functional.log_softmax(preds.reshape(-1, preds.shape[-1]).double(), dim=1)
    target_flat = target.reshape(-1)

    if ignore_index is not None:
        mask = target_flat.ne(ignore_index)
        target_flat = target_flat.where(mask, torch.zeros_like(target_flat))
    else:
        mask = torch.ones_like(target_flat, dtype=torch.bool)

    nll = -probs.gather(1, target_flat.unsqueeze(1)).squeeze(1)
    # accumulate in double for accuracy, report in the input dtype (reference parity)
    total_log_probs = (nll * mask).sum().to(preds.dtype)
    count = mask.sum()
    return total_log_probs, count


def _perplexity_compute(total: Tensor, count: Tensor) -> Tensor:
    return torch.exp(total / count)


def perplexity(preds: Tensor, target: Tensor, ignore_index: Optional[int] = None) -> Tensor:
    """Perplexity from (B, T, V) logits and (B, T) token ids."""
    total, count = _perplexity_update(preds, target, ignore_index)
    return _perplexity_compute(total, count)


# ---------------------------------------------------------- ExtendedEditDistance
def _eed_preprocess(sentence: str, lang: str = "en") -> str:
    """Official EED preprocessing (reference functional/text/eed.py:175 _preprocess_en/_preprocess_ja)."""
    if not isinstance(sentence, str):
        raise ValueError(f"Only strings allowed during preprocessing step, found {type(sentence)} instead")
    if lang == "ja":
        import unicodedata

        return unicodedata.normalize("NFKC", sentence.rstrip())
    sentence = sentence.rstrip()
    for pattern, replacement in ((".", " ."), ("!", " !"), ("?", " ?"), (",", " ,")):
        sentence = sentence.replace(pattern, replacement)
    sentence = re.sub(r"\s+", r" ", sentence)
    sentence = re.sub(r"(\d) ([.,]) (\d)", r"\1\2\3", sentence)  # 0 . 1 -> 0.1
    sentence = re.sub(r"(Dr|Jr|Prof|Rev|Gen|Mr|Mt|Mrs|Ms) .", r"\1.", sentence)  # Mr . -> Mr.
    for pattern, replacement in (("e . g .", "e.g."), ("i . e .", "i.e."), ("U . S .", "U.S.")):
        sentence = sentence.replace(pattern, replacement)
    return " " + sentence + " "


def _eed_single(pred: str, ref: str, alpha: float = 2.0, rho: float = 0.3, deletion: float = 0.2, insertion: float = 1.0) -> float:
    """Extended edit distance (Stanchev et al. 2019), character level with long jumps.

    Follows the official EED DP: the row runs over the HYPOTHESIS, the outer
    loop over the REFERENCE; long jumps (cost ``alpha``) are allowed at blank
    reference positions; ``lj`` counts repeated visits for the coverage term.
    """
    hyp = list(pred)
    ref_ch = list(ref)
    n = len(hyp)

    lj = [-1] * (n + 1)
    row = [1.0] * (n + 1)  # row[i] - edit distance between first i characters of hyp and first w characters of ref
    row[0] = 0.0

    for w in range(1, len(ref_ch) + 1):
        next_row = [float("inf")] * (n + 1)
        for i in range(n + 1):
            if i > 0:
                next_row[i] = min(
                    next_row[i - 1] + deletion,
                    row[i - 1] + (0.0 if ref_ch[w - 1] == hyp[i - 1] else 1.0),
                    row[i] + insertion,
                )
            else:
                next_row[i] = row[i] + 1.0

        min_idx = next_row.index(min(next_row))
        lj[min_idx] += 1
        # long jump at blank reference characters
        if ref_ch[w - 1] == " ":
            jump = alpha + next_row[min_idx]
            next_row = [min(x, jump) for x in next_row]
        row = next_row

    errors = row[n]
    # official EED quirk: unvisited positions (-1) contribute 1 to coverage
    coverage = rho * sum(x if x >= 0 else 1 for x in lj)
    denom = float(len(ref_ch)) + coverage
    return min(1.0, (errors + coverage) / denom) if denom > 0 else 0.0


def extended_edit_distance(
    preds: Union[str, Sequence[str]],
    target: Union[str, Sequence[str], Sequence[Sequence[str]]],
    language: str = "en",
    return_sentence_level_score: bool = False,
    alpha: float = 2.0,
    rho: float = 0.3,
    deletion: float = 0.2,
    insertion: float = 1.0,
):
    """Extended edit distance (lower is better, in [0, 1])."""
    preds_ = [preds] if isinstance(preds, str) else list(preds)
    if isinstance(target, str):
        target_: List[List[str]] = [[target]]
    elif target and isinstance(target[0], str):
        target_ = [[t] for t in target]
    else:
        target_ = [list(t) for t in target]

    scores = []
    for p, refs in zip(preds_, target_):
        p_n = _eed_preprocess(p, language)
        best = min(_eed_single(p_n, _eed_preprocess(r, language), alpha, rho, deletion, insertion) for r in refs)
        scores.append(best)
    scores_t = torch.tensor(scores)
    if return_sentence_level_score:
        return scores_t.mean(), scores_t
    return scores_t.mean()
