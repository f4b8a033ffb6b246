"""Plot API smoke tests (matplotlib available in the test image)."""
import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402
import pytest  # noqa: E402
import torch  # noqa: E402

import metrics_amd as ma  # noqa: E402
from tests.unittests._helpers import seed_all  # noqa: E402

seed_all(81)


def _check(out):
    fig, ax = out
    assert fig is not None and ax is not None
    plt.close(fig)


def test_plot_scalar_metric():
    m = ma.MulticlassAccuracy(num_classes=5)
    m.update(torch.randn(64, 5), torch.randint(0, 5, (64,)))
    _check(m.plot())
    _check(m.plot(val=torch.tensor(0.5)))
    _check(m.plot(val=[torch.tensor(0.4), torch.tensor(0.6)]))  # series


def test_plot_per_class_metric():
    m = ma.MulticlassF1Score(num_classes=5, average=None)
    m.update(torch.randn(64, 5), torch.randint(0, 5, (64,)))
    _check(m.plot())


def test_plot_confusion_matrix():
    m = ma.MulticlassConfusionMatrix(num_classes=4)
    m.update(torch.randint(0, 4, (100,)), torch.randint(0, 4, (100,)))
    _check(m.plot())
    _check(m.plot(add_text=False))


def test_plot_curves():
    m = ma.BinaryPrecisionRecallCurve(thresholds=20)
    m.update(torch.rand(200), torch.randint(0, 2, (200,)))
    _check(m.plot())
    r = ma.BinaryROC(thresholds=20)
    r.update(torch.rand(200), torch.randint(0, 2, (200,)))
    _check(r.plot())
    _check(r.plot(score=True))


def test_plot_collection():
    coll = ma.MetricCollection([ma.BinaryAccuracy(), ma.BinaryF1Score()])
    coll.update(torch.rand(64), torch.randint(0, 2, (64,)))
    figaxs = coll.plot()
    assert isinstance(figaxs, list) and len(figaxs) == 2
    for fig, ax in figaxs:
        plt.close(fig)


def test_plot_tracker():
    tracker = ma.wrappers.MetricTracker(ma.BinaryAccuracy())
    for _ in range(3):
        tracker.increment()
        tracker.update(torch.rand(32), torch.randint(0, 2, (32,)))
    _check(tracker.plot())
