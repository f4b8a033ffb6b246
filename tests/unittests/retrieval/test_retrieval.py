"""Retrieval metric tests vs manual references."""
import numpy as np
import torch

import metrics_amd as ma


IDX = torch.tensor([0, 0, 0, 1, 1, 1, 1])
PREDS = torch.tensor([0.2, 0.3, 0.5, 0.1, 0.3, 0.5, 0.2])
TARGET = torch.tensor([False, False, True, False, True, False, True])


def test_retrieval_map():
    m = ma.RetrievalMAP()
    m.update(PREDS, TARGET, indexes=IDX)
    # q0: ranked [0.5(T), 0.3(F), 0.2(F)] -> AP = 1.0
    # q1: ranked [0.5(F), 0.3(T), 0.2(T), 0.1(F)] -> AP = (1/2 + 2/3)/2
    ref = (1.0 + (0.5 + 2 / 3) / 2) / 2
    assert abs(m.compute().item() - ref) < 1e-6


def test_retrieval_mrr():
    m = ma.RetrievalMRR()
    m.update(PREDS, TARGET, indexes=IDX)
    ref = (1.0 + 0.5) / 2
    assert abs(m.compute().item() - ref) < 1e-6


def test_retrieval_precision_recall_hitrate():
    m = ma.RetrievalPrecision(top_k=2)
    m.update(PREDS, TARGET, indexes=IDX)
    ref = (0.5 + 0.5) / 2  # q0: 1 of top2; q1: 1 of top2
    assert abs(m.compute().item() - ref) < 1e-6

    m = ma.RetrievalRecall(top_k=2)
    m.update(PREDS, TARGET, indexes=IDX)
    ref = (1.0 + 0.5) / 2
    assert abs(m.compute().item() - ref) < 1e-6

    m = ma.RetrievalHitRate(top_k=1)
    m.update(PREDS, TARGET, indexes=IDX)
    ref = (1.0 + 0.0) / 2
    assert abs(m.compute().item() - ref) < 1e-6


def test_retrieval_fallout():
    m = ma.RetrievalFallOut(top_k=2)
    m.update(PREDS, TARGET, indexes=IDX)
    # q0: negatives: 2, in top2: 1 -> 0.5 ; q1: negatives 2, top2 has 1 neg -> 0.5
    assert abs(m.compute().item() - 0.5) < 1e-6


def test_retrieval_ndcg_vs_sklearn():
    from sklearn.metrics import ndcg_score

    torch.manual_seed(3)
    p = torch.rand(30)
    t = torch.randint(0, 4, (30,))
    m = ma.RetrievalNormalizedDCG()
    m.update(p, t, indexes=torch.zeros(30, dtype=torch.long))
    ref = ndcg_score(t[None].numpy(), p[None].numpy())
    assert abs(m.compute().item() - ref) < 1e-5


def test_retrieval_r_precision():
    m = ma.RetrievalRPrecision()
    m.update(PREDS, TARGET, indexes=IDX)
    # q0: R=1, top1 relevant -> 1 ; q1: R=2, top2 has 1 relevant -> 0.5
    assert abs(m.compute().item() - 0.75) < 1e-6


def test_empty_target_actions():
    idx = torch.tensor([0, 0, 1, 1])
    preds = torch.tensor([0.3, 0.2, 0.5, 0.4])
    target = torch.tensor([0, 0, 1, 0])
    for action, expected in (("neg", 0.5), ("pos", 1.0), ("skip", 1.0)):
        m = ma.RetrievalMAP(empty_target_action=action)
        m.update(preds, target, indexes=idx)
        assert abs(m.compute().item() - expected) < 1e-6, action

    import pytest

    m = ma.RetrievalMAP(empty_target_action="error")
    m.update(preds, target, indexes=idx)
    with pytest.raises(ValueError, match="no positive target"):
        m.compute()


def test_retrieval_curve_and_fixed_precision():
    m = ma.RetrievalPrecisionRecallCurve(max_k=3)
    m.update(PREDS, TARGET, indexes=IDX)
    p, r, k = m.compute()
    assert p.shape == (3,) and r.shape == (3,) and (k == torch.tensor([1, 2, 3])).all()

    m = ma.RetrievalRecallAtFixedPrecision(min_precision=0.4, max_k=3)
    m.update(PREDS, TARGET, indexes=IDX)
    best_recall, best_k = m.compute()
    assert 0 <= best_recall <= 1
