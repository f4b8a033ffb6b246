"""Automated state-layout audit vs the reference.

For every same-named Metric class in both top-level namespaces that can be
instantiated with a small common-argument guess, the registered state names
and default shapes/dtypes must match — this is what makes checkpoints
(state_dict with persistent=True) interchangeable across the two
implementations for the whole surface, not just the hand-picked five in
test_state_dict_compat.py.

Known deliberate deviations are listed with reasons.
"""
from __future__ import annotations

import os
import sys

import pytest
import torch

_REF = "/root/reference/src"
HAVE_REF = os.path.isdir(_REF)
pytestmark = pytest.mark.skipif(not HAVE_REF, reason="reference tree not available")

if HAVE_REF:
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "..", "tools", "refbench"))
    sys.path.insert(0, _REF)

import metrics_amd as ma

# argument guesses tried in order until one constructs BOTH classes
_ARG_GUESSES = (
    {},
    {"num_classes": 5},
    {"num_labels": 4},
    {"num_classes": 5, "thresholds": 20},
    {"num_labels": 4, "thresholds": 20},
    {"thresholds": 20},
    {"num_outputs": 1},
    {"task": "binary"},
)

# our state layout deliberately differs (documented MI355X design choices)
_KNOWN_DEVIATIONS = {
    # fused per-class counter layout: micro-average metrics keep (C,) counts
    # so they can share a compute group with macro metrics (see
    # test_ref_differential_modular.test_collection_compute_groups_match)
    "MulticlassAccuracy", "MulticlassPrecision", "MulticlassRecall", "MulticlassF1Score",
    "MulticlassFBetaScore", "MulticlassSpecificity", "MulticlassNegativePredictiveValue",
    "MulticlassHammingDistance", "MulticlassStatScores", "MulticlassJaccardIndex",
    "MulticlassExactMatch",
    # detection: own RLE pack instead of pycocotools tuples
    "MeanAveragePrecision",
    # O(1) accumulated statistics instead of the reference's raw preds/target
    # cat-lists (SURVEY 5.7 state-size handling: no unbounded image buffers)
    "ErrorRelativeGlobalDimensionlessSynthesis", "RelativeAverageSpectralError",
    # simple RCCL-reducible denominator stats (sum/min/max) instead of the
    # reference's running mean/var with dist_reduce_fx=None merge
    "NormalizedRootMeanSquaredError",
    # nominal association metrics: O(1) (C,C) contingency counts instead of
    # the reference's unbounded preds/target cat lists (num_classes is a
    # required argument in both; value parity in test_ref_differential_modular)
    "CramersV", "PearsonsContingencyCoefficient", "TheilsU", "TschuprowsT",
}


def _instantiate(cls):
    for kw in _ARG_GUESSES:
        try:
            return cls(**kw), kw
        except Exception:
            continue
    return None, None


def test_state_layout_matches_reference():
    import torchmetrics as tm

    common = sorted(set(dir(ma)) & set(dir(tm)))
    checked, mismatches, skipped = [], [], []
    for name in common:
        ours_cls = getattr(ma, name)
        ref_cls = getattr(tm, name)
        if not (isinstance(ours_cls, type) and issubclass(ours_cls, ma.Metric)):
            continue
        if name in _KNOWN_DEVIATIONS:
            continue
        import warnings

        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            ours, kw1 = _instantiate(ours_cls)
            ref, kw2 = _instantiate(ref_cls)
        if ours is None or ref is None or kw1 != kw2:
            skipped.append(name)
            continue
        k1, k2 = set(ours._defaults), set(ref._defaults)
        if k1 != k2:
            mismatches.append((name, sorted(k1), sorted(k2)))
            continue
        for k in k1:
            d1, d2 = ours._defaults[k], ref._defaults[k]
            if isinstance(d1, torch.Tensor) != isinstance(d2, torch.Tensor):
                mismatches.append((name, k, type(d1), type(d2)))
            elif isinstance(d1, torch.Tensor) and (d1.shape != d2.shape or d1.dtype != d2.dtype):
                mismatches.append((name, k, (d1.shape, d1.dtype), (d2.shape, d2.dtype)))
        checked.append(name)
    assert len(checked) >= 60, f"audit only covered {len(checked)} classes (skipped: {skipped[:20]})"
    assert not mismatches, mismatches
