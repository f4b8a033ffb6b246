"""Tercom (TER) core — faithful reimplementation of the published algorithm.

Follows the tercom/sacrebleu algorithm as specified by the reference
(torchmetrics functional/text/ter.py + the `_LevenshteinEditDistance`
machinery in functional/text/helper.py), including its observable quirks:

- beam-banded Levenshtein DP (band 25 around the length-ratio pseudo
  diagonal) with tercom's op preference (match/substitute, then delete,
  then insert — pre-flipped order),
- shift candidates limited to spans <= 10 long, start-distance <= 50,
  capped at 1000 tried candidates overall,
- shifts only allowed when both sides are misaligned and the span is not
  shifted into itself; candidate ranking by (gain, length, -earliest pred
  start, -earliest target insert),
- per-reference loop with SWAPPED arguments (the reference shifts the
  target toward the hypothesis) and edits==0 for an empty "target".
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

_BEAM = 25
_MAX_SHIFT_SIZE = 10
_MAX_SHIFT_DIST = 50
_MAX_CANDIDATES = 1000
_INF = int(1e16)

# op codes: 0 nothing, 1 substitute, 2 insert, 3 delete
_NOTHING, _SUB, _INS, _DEL = 0, 1, 2, 3


def _edit_matrix(pred: List[str], ref: List[str]) -> Tuple[int, List[int]]:
    """Beam-banded Levenshtein; returns (distance, flat (cost, op) matrix)."""
    n, m = len(pred), len(ref)
    w = m + 1
    cost = [_INF] * ((n + 1) * w)
    op = [_INS] * ((n + 1) * w)
    for j in range(w):
        cost[j] = j
        op[j] = _INS
    length_ratio = m / n if pred else 1.0
    beam = math.ceil(length_ratio / 2 + _BEAM) if length_ratio / 2 > _BEAM else _BEAM
    for i in range(1, n + 1):
        pseudo_diag = math.floor(i * length_ratio)
        min_j = max(0, pseudo_diag - beam)
        max_j = w if i == n else min(w, pseudo_diag + beam)
        base = i * w
        prev = base - w
        for j in range(min_j, max_j):
            if j == 0:
                cost[base] = cost[prev] + 1
                op[base] = _DEL
                continue
            if pred[i - 1] == ref[j - 1]:
                c_sub, o_sub = cost[prev + j - 1], _NOTHING
            else:
                c_sub, o_sub = cost[prev + j - 1] + 1, _SUB
            best_c, best_o = c_sub, o_sub
            c = cost[prev + j] + 1
            if best_c > c:
                best_c, best_o = c, _DEL
            c = cost[base + j - 1] + 1
            if best_c > c:
                best_c, best_o = c, _INS
            cost[base + j] = best_c
            op[base + j] = best_o
    return cost[n * w + m], op


def _trace(pred_len: int, ref_len: int, op: List[int]) -> List[int]:
    w = ref_len + 1
    i, j = pred_len, ref_len
    out: List[int] = []
    while i > 0 or j > 0:
        o = op[i * w + j]
        out.append(o)
        if o in (_NOTHING, _SUB):
            i -= 1
            j -= 1
        elif o == _INS:
            j -= 1
        else:
            i -= 1
    out.reverse()
    return out


def _alignment(trace: List[int]) -> Tuple[Dict[int, int], List[int], List[int]]:
    """Flipped-trace alignment: maps ref positions to hyp positions + errors."""
    # flipping swaps INS <-> DEL; fold the flip into the walk
    ref_pos = hyp_pos = -1
    ref_err: List[int] = []
    hyp_err: List[int] = []
    align: Dict[int, int] = {}
    for o in trace:
        if o == _NOTHING:
            hyp_pos += 1
            ref_pos += 1
            align[ref_pos] = hyp_pos
            ref_err.append(0)
            hyp_err.append(0)
        elif o == _SUB:
            hyp_pos += 1
            ref_pos += 1
            align[ref_pos] = hyp_pos
            ref_err.append(1)
            hyp_err.append(1)
        elif o == _DEL:  # flipped: acts as INSERT on the hyp side
            hyp_pos += 1
            hyp_err.append(1)
        else:  # _INS flipped: DELETE — consumes a ref word
            ref_pos += 1
            align[ref_pos] = hyp_pos
            ref_err.append(1)
    return align, ref_err, hyp_err


def _do_shift(words: List[str], start: int, length: int, to: int) -> List[str]:
    if to < start:
        return words[:to] + words[start : start + length] + words[to:start] + words[start + length :]
    if to > start + length:
        return words[:start] + words[start + length : to] + words[start : start + length] + words[to:]
    return (
        words[:start]
        + words[start + length : length + to]
        + words[start : start + length]
        + words[length + to :]
    )


def _best_shift(
    pred: List[str], ref: List[str], base_dist: int, checked: int
) -> Tuple[int, List[str], int]:
    _, op = _edit_matrix(pred, ref)
    align, ref_err, hyp_err = _alignment(_trace(len(pred), len(ref), op))

    best: Optional[Tuple[int, int, int, int, List[str]]] = None
    np_, nr = len(pred), len(ref)
    for ps in range(np_):
        for ts in range(nr):
            if abs(ts - ps) > _MAX_SHIFT_DIST:
                continue
            for length in range(1, _MAX_SHIFT_SIZE):
                if ps + length > np_ or ts + length > nr:
                    break
                if pred[ps + length - 1] != ref[ts + length - 1]:
                    break
                # the span must be wrong on the hyp side AND at the target site,
                # and must not be shifted into itself
                ok = (
                    sum(hyp_err[ps : ps + length]) != 0
                    and sum(ref_err[ts : ts + length]) != 0
                    and not (ps <= align[ts] < ps + length)
                )
                if ok:
                    prev_idx = -1
                    for off in range(-1, length):
                        if ts + off == -1:
                            idx = 0
                        elif ts + off in align:
                            idx = align[ts + off] + 1
                        else:
                            break
                        if idx == prev_idx:
                            continue
                        prev_idx = idx
                        cand_words = _do_shift(pred, ps, length, idx)
                        d, _ = _edit_matrix(cand_words, ref)
                        cand = (base_dist - d, length, -ps, -idx, cand_words)
                        checked += 1
                        if best is None or cand > best:
                            best = cand
                if ps + length >= np_ or ts + length >= nr:
                    break
                if checked >= _MAX_CANDIDATES:
                    break
            if checked >= _MAX_CANDIDATES:
                break
        if checked >= _MAX_CANDIDATES:
            break
    if best is None:
        return 0, pred, checked
    gain, _, _, _, shifted = best
    return gain, shifted, checked


def tercom_edits(pred_words: List[str], target_words: List[str]) -> float:
    """Number of tercom edits (shifts + remaining edit distance)."""
    if len(target_words) == 0:
        return 0.0
    words = pred_words
    shifts = 0
    checked = 0
    while True:
        base, _ = _edit_matrix(words, target_words)
        gain, new_words, checked = _best_shift(words, target_words, base, checked)
        if checked >= _MAX_CANDIDATES or gain <= 0:
            break
        shifts += 1
        words = new_words
    dist, _ = _edit_matrix(words, target_words)
    return float(shifts + dist)


def sentence_ter(pred_words: List[str], refs_words: List[List[str]]) -> Tuple[float, float]:
    """(best edit count, average reference length) for one hypothesis.

    Reference quirk kept: the per-reference call swaps the roles, shifting
    the REFERENCE toward the hypothesis.
    """
    total_len = 0.0
    best = float(2e16)
    for ref in refs_words:
        edits = tercom_edits(ref, pred_words)
        total_len += len(ref)
        best = min(best, edits)
    return best, total_len / len(refs_words)
