"""Nominal association metrics: known values + scipy contingency oracles."""
import numpy as np
import pytest
import torch
from scipy.stats import chi2_contingency
from scipy.stats.contingency import association

import metrics_amd as ma
from tests.unittests._helpers import seed_all

seed_all(72)
N = 500
X = torch.randint(0, 5, (N,))
Y = (X + torch.randint(0, 3, (N,))) % 5  # correlated


def _confmat(x, y, k=5):
    cm = np.zeros((k, k), dtype=np.int64)
    for a, b in zip(x.tolist(), y.tolist()):
        cm[a, b] += 1
    return cm


def test_cramers_v_vs_scipy():
    # scipy's correction=True is Yates continuity, NOT the Bergsma bias
    # correction the reference implements — compare uncorrected values only
    ref_nc = association(_confmat(X, Y), method="cramer", correction=False)
    m2 = ma.CramersV(num_classes=5, bias_correction=False)
    m2.update(X, Y)
    assert abs(m2.compute().item() - ref_nc) < 1e-5
    m = ma.CramersV(num_classes=5, bias_correction=True)
    m.update(X, Y)
    corrected = m.compute().item()
    assert 0 <= corrected <= ref_nc + 1e-9  # Bergsma correction shrinks toward 0


def test_tschuprows_t_vs_scipy():
    ref = association(_confmat(X, Y), method="tschuprow", correction=False)
    m = ma.TschuprowsT(num_classes=5, bias_correction=False)
    m.update(X, Y)
    assert abs(m.compute().item() - ref) < 1e-5


def test_pearsons_contingency_vs_scipy():
    ref = association(_confmat(X, Y), method="pearson")
    m = ma.PearsonsContingencyCoefficient(num_classes=5)
    m.update(X, Y)
    assert abs(m.compute().item() - ref) < 1e-5


def test_theils_u_bounds_and_direction():
    m = ma.TheilsU(num_classes=5)
    m.update(X, Y)
    v = m.compute().item()
    assert 0 < v < 1
    # deterministic relation -> U(X|X) == 1
    m2 = ma.TheilsU(num_classes=5)
    m2.update(X, X)
    assert abs(m2.compute().item() - 1.0) < 1e-6


def test_fleiss_kappa_vs_manual():
    from metrics_amd.functional.nominal import fleiss_kappa

    seed_all(74)
    # (subjects, categories) counts with 6 raters each
    ratings = torch.zeros(30, 4, dtype=torch.long)
    for i in range(30):
        picks = torch.randint(0, 4, (6,))
        for p_ in picks.tolist():
            ratings[i, p_] += 1
    m = ratings.double().numpy()
    n = m.sum(1)[0]
    p_i = ((m**2).sum(1) - n) / (n * (n - 1))
    pj = m.sum(0) / m.sum()
    kappa_ref = (p_i.mean() - (pj**2).sum()) / (1 - (pj**2).sum())
    assert abs(fleiss_kappa(ratings, mode="counts").item() - kappa_ref) < 1e-4
