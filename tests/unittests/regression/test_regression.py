"""sklearn/scipy-oracle tests for the regression domain."""
import numpy as np
import pytest
import scipy.stats
import torch
from sklearn import metrics as skm

import metrics_amd as ma
from tests.unittests._helpers import run_class_metric_test, seed_all


@pytest.fixture()
def data():
    seed_all(21)
    return torch.randn(4, 48), torch.randn(4, 48)


@pytest.fixture()
def data_pos():
    seed_all(22)
    return torch.rand(4, 48) + 0.1, torch.rand(4, 48) + 0.1


def test_mse(data):
    preds, target = data
    run_class_metric_test(ma.MeanSquaredError, lambda p, t: skm.mean_squared_error(t.numpy(), p.numpy()), preds, target)


def test_rmse(data):
    preds, target = data
    run_class_metric_test(
        ma.MeanSquaredError,
        lambda p, t: np.sqrt(skm.mean_squared_error(t.numpy(), p.numpy())),
        preds, target, {"squared": False},
    )


def test_mse_multioutput():
    seed_all(23)
    preds, target = torch.randn(4, 32, 3), torch.randn(4, 32, 3)
    run_class_metric_test(
        ma.MeanSquaredError,
        lambda p, t: skm.mean_squared_error(t.reshape(-1, 3).numpy(), p.reshape(-1, 3).numpy(), multioutput="raw_values"),
        preds, target, {"num_outputs": 3},
    )


def test_mae(data):
    preds, target = data
    run_class_metric_test(ma.MeanAbsoluteError, lambda p, t: skm.mean_absolute_error(t.numpy(), p.numpy()), preds, target)


def test_mape(data_pos):
    preds, target = data_pos
    run_class_metric_test(
        ma.MeanAbsolutePercentageError,
        lambda p, t: skm.mean_absolute_percentage_error(t.numpy(), p.numpy()),
        preds, target, atol=1e-4,
    )


def test_smape(data_pos):
    preds, target = data_pos

    def ref(p, t):
        p, t = p.numpy(), t.numpy()
        return np.mean(2 * np.abs(p - t) / (np.abs(p) + np.abs(t)))

    run_class_metric_test(ma.SymmetricMeanAbsolutePercentageError, ref, preds, target, atol=1e-4)


def test_wmape(data_pos):
    preds, target = data_pos

    def ref(p, t):
        p, t = p.numpy(), t.numpy()
        return np.abs(p - t).sum() / np.abs(t).sum()

    run_class_metric_test(ma.WeightedMeanAbsolutePercentageError, ref, preds, target, atol=1e-4)


def test_msle(data_pos):
    preds, target = data_pos
    run_class_metric_test(
        ma.MeanSquaredLogError, lambda p, t: skm.mean_squared_log_error(t.numpy(), p.numpy()), preds, target
    )


def test_r2(data):
    preds, target = data
    run_class_metric_test(
        ma.R2Score, lambda p, t: skm.r2_score(t.numpy(), p.numpy()), preds, target, check_batch=False
    )


def test_explained_variance(data):
    preds, target = data
    run_class_metric_test(
        ma.ExplainedVariance,
        lambda p, t: skm.explained_variance_score(t.numpy(), p.numpy()),
        preds, target, check_batch=False,
    )


def test_pearson(data):
    preds, target = data
    # correlated inputs
    target = preds * 0.5 + 0.3 * target
    run_class_metric_test(
        ma.PearsonCorrCoef,
        lambda p, t: scipy.stats.pearsonr(t.numpy().flatten(), p.numpy().flatten())[0],
        preds, target, check_batch=False, atol=1e-4,
    )


def test_spearman(data):
    preds, target = data
    target = preds * 0.5 + 0.3 * target
    run_class_metric_test(
        ma.SpearmanCorrCoef,
        lambda p, t: scipy.stats.spearmanr(t.numpy().flatten(), p.numpy().flatten())[0],
        preds, target, check_batch=False, atol=1e-4,
    )


def test_kendall(data):
    preds, target = data
    target = preds * 0.5 + 0.3 * target
    run_class_metric_test(
        ma.KendallRankCorrCoef,
        lambda p, t: scipy.stats.kendalltau(t.numpy().flatten(), p.numpy().flatten())[0],
        preds, target, check_batch=False, atol=1e-4,
    )


def test_concordance(data):
    preds, target = data
    target = preds * 0.5 + 0.3 * target

    def ref(p, t):
        p, t = p.numpy().flatten(), t.numpy().flatten()
        r = scipy.stats.pearsonr(t, p)[0]
        return 2 * r * p.std() * t.std() / (p.var() + t.var() + (p.mean() - t.mean()) ** 2)

    run_class_metric_test(ma.ConcordanceCorrCoef, ref, preds, target, check_batch=False, atol=1e-4)


def test_cosine_similarity():
    seed_all(24)
    preds, target = torch.randn(4, 16, 8), torch.randn(4, 16, 8)

    def ref(p, t):
        p, t = p.reshape(-1, 8).numpy(), t.reshape(-1, 8).numpy()
        sims = (p * t).sum(-1) / (np.linalg.norm(p, axis=-1) * np.linalg.norm(t, axis=-1))
        return sims.mean()

    run_class_metric_test(ma.CosineSimilarity, ref, preds, target, {"reduction": "mean"}, atol=1e-5)


def test_kl_divergence():
    seed_all(25)
    preds = torch.softmax(torch.randn(4, 16, 10), -1)
    target = torch.softmax(torch.randn(4, 16, 10), -1)

    def ref(p, t):
        p, t = p.reshape(-1, 10).numpy(), t.reshape(-1, 10).numpy()
        return (p * np.log(p / t)).sum(-1).mean()

    run_class_metric_test(ma.KLDivergence, ref, preds, target, atol=1e-5)


def test_minkowski(data):
    preds, target = data
    run_class_metric_test(
        ma.MinkowskiDistance,
        lambda p, t: float(np.power(np.power(np.abs(p.numpy() - t.numpy()), 3).sum(), 1 / 3)),
        preds, target, {"p": 3}, check_batch=False, atol=1e-4,
    )


def test_log_cosh(data):
    preds, target = data

    def ref(p, t):
        d = p.numpy() - t.numpy()
        return np.mean(np.log(np.cosh(d)))

    run_class_metric_test(ma.LogCoshError, ref, preds, target, atol=1e-5)


def test_tweedie_deviance(data_pos):
    preds, target = data_pos
    run_class_metric_test(
        ma.TweedieDevianceScore,
        lambda p, t: skm.mean_tweedie_deviance(t.numpy(), p.numpy(), power=1.5),
        preds, target, {"power": 1.5}, atol=1e-4,
    )


def test_relative_squared_error(data):
    preds, target = data

    def ref(p, t):
        p, t = p.numpy(), t.numpy()
        return ((t - p) ** 2).sum() / ((t - t.mean()) ** 2).sum()

    run_class_metric_test(ma.RelativeSquaredError, ref, preds, target, check_batch=False, atol=1e-4)


def test_critical_success_index():
    seed_all(26)
    preds, target = torch.rand(4, 64), torch.rand(4, 64)

    def ref(p, t):
        pb, tb = p.numpy() >= 0.5, t.numpy() >= 0.5
        hits = (pb & tb).sum()
        return hits / (hits + (~pb & tb).sum() + (pb & ~tb).sum())

    run_class_metric_test(ma.CriticalSuccessIndex, ref, preds, target, {"threshold": 0.5})


def test_nrmse(data):
    preds, target = data

    def ref(p, t):
        p, t = p.numpy(), t.numpy()
        return np.sqrt(((p - t) ** 2).mean()) / np.abs(t.mean())

    run_class_metric_test(ma.NormalizedRootMeanSquaredError, ref, preds, target, check_batch=False, atol=1e-4)


def test_pearson_ddp_merge():
    """The (world, ...) stacked-state merge path produces the global pearson."""
    from tests.unittests._helpers import run_distributed

    def _worker(rank, world):
        seed_all(30)
        preds = torch.randn(4, 50)
        target = preds * 0.5 + 0.3 * torch.randn(4, 50)
        m = ma.PearsonCorrCoef()
        for i in range(rank, 4, world):
            m.update(preds[i], target[i])
        v = m.compute()
        ref = scipy.stats.pearsonr(target.flatten().numpy(), preds.flatten().numpy())[0]
        assert abs(v.item() - ref) < 1e-4, (v.item(), ref)

    run_distributed(_worker, world_size=2)


def test_reference_doctest_values_cross_domain():
    """Values pinned to reference doctests (same operand order)."""
    import metrics_amd as ma
    from metrics_amd import functional as F

    idx = torch.tensor([0, 0, 0, 1, 1, 1, 1])
    preds = torch.tensor([0.2, 0.3, 0.5, 0.1, 0.3, 0.5, 0.2])
    target = torch.tensor([False, False, True, False, True, False, True])
    m = ma.RetrievalMRR(); m.update(preds, target, indexes=idx)
    assert abs(float(m.compute()) - 0.75) < 1e-4
    m = ma.RetrievalMAP(); m.update(preds, target, indexes=idx)
    assert abs(float(m.compute()) - 0.7917) < 1e-4

    assert abs(float(F.cosine_similarity(
        torch.tensor([[0., 1.], [1., 1.]]), torch.tensor([[0., 1.], [0., 1.]]), reduction="mean")) - 0.8536) < 1e-4
    # reference doc: preds=[2.5,0,2,8], target=[3,-0.5,2,7] -> 0.9572
    assert abs(float(F.explained_variance(
        torch.tensor([2.5, 0.0, 2, 8]), torch.tensor([3., -0.5, 2, 7]))) - 0.9572) < 1e-4
    assert abs(float(F.kendall_rank_corrcoef(
        torch.tensor([2., 7, 20, 200]), torch.tensor([0.3, 0.2, 0.6, 0.1]))) + 0.3333) < 1e-4
    assert abs(float(F.tweedie_deviance_score(
        torch.tensor([4.0, 3.0, 2.0, 1.0]), torch.tensor([1.0, 2.0, 3.0, 4.0]), power=2)) - 1.2083) < 1e-4
    assert abs(float(F.mean_squared_log_error(
        torch.tensor([0., 1, 2, 3]), torch.tensor([0., 1, 2, 2]))) - 0.0207) < 1e-4
    assert abs(float(F.symmetric_mean_absolute_percentage_error(
        torch.tensor([1., 10, 1e6]), torch.tensor([0.9, 15, 1.2e6]))) - 0.2290) < 1e-4

    m = ma.RunningMean(window=3)
    for v in [1.0, 2.0, 3.0, 4.0]:
        m.update(torch.tensor(v))
    assert abs(float(m.compute()) - 3.0) < 1e-6
