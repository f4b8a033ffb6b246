"""RetrievalAUROC + wrapper bits not covered elsewhere."""
import pytest
import torch
from sklearn.metrics import roc_auc_score

import metrics_amd as ma
from tests.unittests._helpers import seed_all


def test_retrieval_auroc_vs_sklearn():
    seed_all(62)
    idx = torch.repeat_interleave(torch.arange(6), 20)
    preds = torch.rand(120)
    target = torch.randint(0, 2, (120,))
    # make sure every group has both classes
    for g in range(6):
        target[g * 20] = 1
        target[g * 20 + 1] = 0
    m = ma.RetrievalAUROC()
    m.update(preds, target, indexes=idx)
    ref = sum(
        roc_auc_score(target[g * 20 : (g + 1) * 20].numpy(), preds[g * 20 : (g + 1) * 20].numpy()) for g in range(6)
    ) / 6
    assert abs(float(m.compute()) - ref) < 1e-6


def test_fleiss_kappa_class():
    seed_all(63)
    ratings = torch.zeros(20, 3, dtype=torch.long)
    for i in range(20):
        picks = torch.randint(0, 3, (5,))
        for p in picks.tolist():
            ratings[i, p] += 1
    m = ma.FleissKappa(mode="counts")
    m.update(ratings)
    from metrics_amd.functional.nominal import fleiss_kappa

    assert torch.allclose(m.compute(), fleiss_kappa(ratings, mode="counts"))


def test_procrustes_class_and_wrappers():
    a = torch.randn(4, 12, 3)
    m = ma.shape.ProcrustesDisparity()
    m.update(a, a)
    assert float(m.compute()) < 1e-10

    # MetricInputTransformer base: subclass transforms inputs
    from metrics_amd.wrappers import MetricInputTransformer

    class Halve(MetricInputTransformer):
        def transform_pred(self, pred):
            return pred / 2

    w = Halve(ma.MeanSquaredError())
    w.update(torch.ones(4) * 2, torch.zeros(4))
    assert abs(float(w.compute()) - 1.0) < 1e-6


def test_empty_target_action_semantics():
    idx = torch.tensor([0, 0, 1, 1])
    preds = torch.tensor([0.9, 0.1, 0.8, 0.2])
    target = torch.tensor([0, 0, 1, 0])  # group 0 has no positives
    for action, expect in [("skip", 1.0), ("neg", 0.5), ("pos", 1.0)]:
        m = ma.RetrievalMRR(empty_target_action=action)
        m.update(preds, target, indexes=idx)
        assert abs(float(m.compute()) - expect) < 1e-6, action
    import pytest as _pytest

    m = ma.RetrievalMRR(empty_target_action="error")
    m.update(preds, target, indexes=idx)
    with _pytest.raises(ValueError):
        m.compute()


@pytest.mark.parametrize(
    ("cls", "kwargs", "graded"),
    [
        ("RetrievalMAP", {}, False),
        ("RetrievalMAP", {"top_k": 5}, False),
        ("RetrievalMRR", {}, False),
        ("RetrievalMRR", {"top_k": 3}, False),
        ("RetrievalPrecision", {"top_k": 4}, False),
        ("RetrievalPrecision", {"top_k": 100}, False),
        ("RetrievalPrecision", {"top_k": 100, "adaptive_k": True}, False),
        ("RetrievalRecall", {"top_k": 4}, False),
        ("RetrievalHitRate", {"top_k": 2}, False),
        ("RetrievalFallOut", {"top_k": 4}, False),
        ("RetrievalNormalizedDCG", {}, True),
        ("RetrievalNormalizedDCG", {"top_k": 6}, True),
        ("RetrievalRPrecision", {}, False),
    ],
)
def test_batched_retrieval_matches_per_query_loop(cls, kwargs, graded):
    """The vectorized all-queries path equals the per-query loop exactly."""
    import metrics_amd.retrieval as R

    torch.manual_seed(11)
    n, q = 3000, 200
    idx = torch.randint(0, q, (n,))
    preds = torch.rand(n)
    target = torch.randint(0, 4 if graded else 2, (n,))
    klass = getattr(R, cls)
    fast = klass(**kwargs)
    fast.update(preds, target, indexes=idx)
    loop = klass(**kwargs)
    loop._batched_scores = lambda g: None  # force the fallback loop
    loop.update(preds, target, indexes=idx)
    assert torch.allclose(fast.compute(), loop.compute(), atol=1e-6)


@pytest.mark.parametrize("action", ["neg", "pos", "skip"])
def test_batched_retrieval_empty_target_actions(action):
    """empty_target_action handling matches between batched and loop paths."""
    import metrics_amd.retrieval as R

    idx = torch.tensor([0, 0, 1, 1, 2, 2])
    preds = torch.tensor([0.9, 0.2, 0.4, 0.3, 0.8, 0.1])
    target = torch.tensor([1, 0, 0, 0, 1, 1])  # query 1 has no positives
    fast = R.RetrievalMAP(empty_target_action=action)
    fast.update(preds, target, indexes=idx)
    loop = R.RetrievalMAP(empty_target_action=action)
    loop._batched_scores = lambda g: None
    loop.update(preds, target, indexes=idx)
    assert torch.allclose(fast.compute(), loop.compute())

    err = R.RetrievalMAP(empty_target_action="error")
    err.update(preds, target, indexes=idx)
    with pytest.raises(ValueError, match="no positive"):
        err.compute()
