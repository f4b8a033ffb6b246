"""Clustering metrics vs sklearn oracles."""
import pytest
import torch
from sklearn import metrics as skm

import metrics_amd as ma
from tests.unittests._helpers import seed_all

seed_all(71)
N = 300
PREDS = torch.randint(0, 8, (N,))
TARGET = torch.randint(0, 6, (N,))
DATA = torch.randn(N, 5)
DLABELS = torch.randint(0, 4, (N,))


@pytest.mark.parametrize(
    ("metric_cls", "sk_fn"),
    [
        (ma.clustering.MutualInfoScore, skm.mutual_info_score),
        (ma.clustering.AdjustedMutualInfoScore, skm.adjusted_mutual_info_score),
        (ma.clustering.NormalizedMutualInfoScore, skm.normalized_mutual_info_score),
        (ma.clustering.RandScore, skm.rand_score),
        (ma.clustering.AdjustedRandScore, skm.adjusted_rand_score),
        (ma.clustering.FowlkesMallowsIndex, skm.fowlkes_mallows_score),
        (ma.clustering.HomogeneityScore, skm.homogeneity_score),
        (ma.clustering.CompletenessScore, skm.completeness_score),
        (ma.clustering.VMeasureScore, skm.v_measure_score),
    ],
)
def test_label_clustering_vs_sklearn(metric_cls, sk_fn):
    m = metric_cls()
    # chunked updates must accumulate to the full-data value (cat states)
    m.update(PREDS[:150], TARGET[:150])
    m.update(PREDS[150:], TARGET[150:])
    ref = sk_fn(TARGET.numpy(), PREDS.numpy())
    assert abs(m.compute().item() - ref) < 1e-5, metric_cls.__name__


@pytest.mark.parametrize(
    ("metric_cls", "sk_fn"),
    [
        (ma.clustering.CalinskiHarabaszScore, skm.calinski_harabasz_score),
        (ma.clustering.DaviesBouldinScore, skm.davies_bouldin_score),
    ],
)
def test_intrinsic_clustering_vs_sklearn(metric_cls, sk_fn):
    m = metric_cls()
    m.update(DATA[:100], DLABELS[:100])
    m.update(DATA[100:], DLABELS[100:])
    ref = sk_fn(DATA.numpy(), DLABELS.numpy())
    assert abs(m.compute().item() - ref) / max(abs(ref), 1) < 1e-4, metric_cls.__name__


def test_dunn_index():
    x = torch.tensor([[0.0, 0.0], [0.1, 0.0], [5.0, 5.0], [5.1, 5.0]])
    labels = torch.tensor([0, 0, 1, 1])
    v = ma.clustering.DunnIndex()(x, labels).item()
    # min inter-cluster (centroid-based per torchmetrics impl) >> max intra diameter
    assert v > 1
