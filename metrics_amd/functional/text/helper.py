"""Text helpers: Levenshtein DP + tokenizers.

Parity: torchmetrics ``functional/text/helper.py``.
"""
from __future__ import annotations

import re
from typing import List, Sequence, Tuple, Union


def _intern_pairs(pairs):
    """Map token sequences to int32 id arrays + offsets for the native batch.

    Plain strings (char-level CER/EditDistance) skip interning: their
    utf-32-le byte view IS the int32 codepoint array.
    """
    import numpy as np

    ids: dict = {}

    def _to_ids(seq):
        if isinstance(seq, str):
            return np.frombuffer(seq.encode("utf-32-le"), dtype=np.int32)
        out = np.empty(len(seq), dtype=np.int32)
        for i, tok in enumerate(seq):
            out[i] = ids.setdefault(tok, len(ids))
        return out

    a_arrs = [_to_ids(p) for p, _ in pairs]
    b_arrs = [_to_ids(t) for _, t in pairs]
    off_a = np.zeros(len(pairs) + 1, dtype=np.int64)
    off_b = np.zeros(len(pairs) + 1, dtype=np.int64)
    off_a[1:] = np.cumsum([len(x) for x in a_arrs])
    off_b[1:] = np.cumsum([len(x) for x in b_arrs])
    tok_a = np.concatenate(a_arrs) if a_arrs else np.zeros(0, np.int32)
    tok_b = np.concatenate(b_arrs) if b_arrs else np.zeros(0, np.int32)
    return tok_a.astype(np.int32), off_a, tok_b.astype(np.int32), off_b


def _edit_distance_batch(pairs) -> list:
    """Levenshtein distances for a batch of (pred_tokens, ref_tokens) pairs.

    Uses the OpenMP native kernel (csrc/edit_distance.cpp) when the CPU
    library is built; falls back to the Python DP otherwise.
    """
    from metrics_amd.ops import _coco

    if not pairs:
        return []
    lib = _coco._load()
    if lib is None:
        return [_edit_distance(list(p), list(t)) for p, t in pairs]
    import ctypes

    import numpy as np

    tok_a, off_a, tok_b, off_b = _intern_pairs(pairs)
    out = np.zeros(len(pairs), dtype=np.int64)
    lib.ma_edit_distance_batch(
        _coco._ptr(tok_a), _coco._ptr(off_a), _coco._ptr(tok_b), _coco._ptr(off_b),
        ctypes.c_int64(len(pairs)), ctypes.c_int(0), _coco._ptr(out),
    )
    return out.tolist()


def _edit_distance_counts_batch(pairs) -> list:
    """(subs, ins, dels, hits) per pair, native when available."""
    from metrics_amd.ops import _coco

    if not pairs:
        return []
    lib = _coco._load()
    if lib is None:
        return [_edit_distance_counts(p, t) for p, t in pairs]
    import ctypes

    import numpy as np

    tok_a, off_a, tok_b, off_b = _intern_pairs(pairs)
    out = np.zeros(4 * len(pairs), dtype=np.int64)
    lib.ma_edit_distance_batch(
        _coco._ptr(tok_a), _coco._ptr(off_a), _coco._ptr(tok_b), _coco._ptr(off_b),
        ctypes.c_int64(len(pairs)), ctypes.c_int(1), _coco._ptr(out),
    )
    return [tuple(out[4 * i : 4 * i + 4].tolist()) for i in range(len(pairs))]


def _edit_distance(prediction_tokens: Sequence, reference_tokens: Sequence) -> int:
    """Levenshtein distance between two token sequences (O(nm) DP, two rows)."""
    n, m = len(prediction_tokens), len(reference_tokens)
    if n == 0:
        return m
    if m == 0:
        return n
    prev = list(range(m + 1))
    cur = [0] * (m + 1)
    for i in range(1, n + 1):
        cur[0] = i
        p = prediction_tokens[i - 1]
        for j in range(1, m + 1):
            cost = 0 if p == reference_tokens[j - 1] else 1
            cur[j] = min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + cost)
        prev, cur = cur, prev
    return prev[m]


def _edit_distance_counts(prediction_tokens: Sequence, reference_tokens: Sequence) -> Tuple[int, int, int, int]:
    """(substitutions, insertions, deletions, matches) via full DP backtrace."""
    n, m = len(prediction_tokens), len(reference_tokens)
    dp = [[0] * (m + 1) for _ in range(n + 1)]
    for i in range(n + 1):
        dp[i][0] = i
    for j in range(m + 1):
        dp[0][j] = j
    for i in range(1, n + 1):
        for j in range(1, m + 1):
            cost = 0 if prediction_tokens[i - 1] == reference_tokens[j - 1] else 1
            dp[i][j] = min(dp[i - 1][j] + 1, dp[i][j - 1] + 1, dp[i - 1][j - 1] + cost)
    # backtrace
    i, j = n, m
    subs = ins = dels = hits = 0
    while i > 0 or j > 0:
        if i > 0 and j > 0 and dp[i][j] == dp[i - 1][j - 1] + (0 if prediction_tokens[i - 1] == reference_tokens[j - 1] else 1):
            if prediction_tokens[i - 1] == reference_tokens[j - 1]:
                hits += 1
            else:
                subs += 1
            i -= 1
            j -= 1
        elif i > 0 and dp[i][j] == dp[i - 1][j] + 1:
            ins += 1  # extra token in prediction
            i -= 1
        else:
            dels += 1  # missing token from reference
            j -= 1
    return subs, ins, dels, hits


_13A_RE1 = re.compile(r"([\{-\~\[-\` -\&\(-\+\:-\@\/])")
_13A_RE_NUM = re.compile(r"([0-9])([\.,])")
_13A_RE_NUM2 = re.compile(r"([\.,])([0-9])")
_13A_RE_DASH = re.compile(r"([0-9])(-)")


# sacrebleu's mteval-equivalent post-processing rules
_SB_REGEX = (
    (re.compile(r"([\{-\~\[-\` -\&\(-\+\:-\@\/])"), r" \1 "),   # symbols/punct blocks
    (re.compile(r"([^0-9])([\.,])"), r"\1 \2 "),                 # . , not after digit
    (re.compile(r"([\.,])([^0-9])"), r" \1 \2"),                 # . , not before digit
    (re.compile(r"([0-9])(-)"), r"\1 \2 "),                      # dash after digit
)

# CJK blocks the zh tokenizer pads (sacrebleu's published range table)
_CJK_RANGES = (
    ("㐀", "䶵"), ("一", "龥"), ("龦", "龻"),
    ("豈", "鶴"), ("侮", "頻"), ("並", "龎"),
    ("\U00020000", "\U0002a6d6"), ("\U0002f800", "\U0002fa1d"),
    ("＀", "￯"), ("⺀", "⻿"), ("　", "〿"),
    ("㇀", "㇯"), ("⼀", "⿟"), ("⿰", "⿿"),
    ("㄀", "ㄯ"), ("ㆠ", "ㆿ"), ("︐", "︟"),
    ("︰", "﹏"), ("☀", "⛿"), ("✀", "➿"),
    ("㈀", "㋿"), ("㌀", "㏿"),
)


def _sb_regex_post(line: str) -> str:
    for rx, repl in _SB_REGEX:
        line = rx.sub(repl, line)
    return " ".join(line.split())


def _finish(line: str, lowercase: bool) -> List[str]:
    # lowercase applies AFTER tokenization (reference __call__ order)
    return (line.lower() if lowercase else line).split()


def _tokenize_13a(line: str, lowercase: bool = False) -> List[str]:
    """mteval-v13a tokenizer (sacrebleu default)."""
    line = line.replace("<skipped>", "")
    line = line.replace("-\n", "")
    line = line.replace("\n", " ")
    if "&" in line:
        line = line.replace("&quot;", '"').replace("&amp;", "&").replace("&lt;", "<").replace("&gt;", ">")
    return _finish(_sb_regex_post(f" {line} "), lowercase)


def _tokenize_intl(line: str, lowercase: bool = False) -> List[str]:
    """International tokenizer: unicode punctuation/symbol splitting (regex module)."""
    import regex

    rules = getattr(_tokenize_intl, "_rules", None)
    if rules is None:
        rules = (
            (regex.compile(r"(\P{N})(\p{P})"), r"\1 \2 "),
            (regex.compile(r"(\p{P})(\P{N})"), r" \1 \2"),
            (regex.compile(r"(\p{S})"), r" \1 "),
        )
        _tokenize_intl._rules = rules
    for rx, repl in rules:
        line = rx.sub(repl, line)
    return _finish(line, lowercase)


def _tokenize_char(line: str, lowercase: bool = False) -> List[str]:
    # every non-whitespace character is a token (spaces become separators)
    return _finish(" ".join(ch for ch in line), lowercase)


def _tokenize_zh(line: str, lowercase: bool = False) -> List[str]:
    def is_cjk(ch: str) -> bool:
        return any(lo <= ch <= hi for lo, hi in _CJK_RANGES)

    line = line.strip()
    padded = "".join(f" {ch} " if is_cjk(ch) else ch for ch in line)
    return _finish(_sb_regex_post(padded), lowercase)


def _tokenize_none(line: str, lowercase: bool = False) -> List[str]:
    return _finish(line, lowercase)


_TOKENIZERS = {
    "13a": _tokenize_13a,
    "char": _tokenize_char,
    "none": _tokenize_none,
    "intl": _tokenize_intl,
    "zh": _tokenize_zh,
}


def get_tokenizer(name: str):
    if name in ("ja-mecab", "ko-mecab", "flores101", "flores200"):
        raise ModuleNotFoundError(
            f"Tokenizer `{name}` needs an external model/dependency that is not available offline"
        )
    if name not in _TOKENIZERS:
        raise ValueError(f"Unsupported tokenizer {name}; expected one of {list(_TOKENIZERS)}")
    return _TOKENIZERS[name]
