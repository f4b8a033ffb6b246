"""Additional core-behavior coverage: compute_on_cpu, set_dtype on real
metrics, full-state-property checker, merge_state on real metrics, device
moves, persistence round-trips through nn.Module containers."""
import pickle

import pytest
import torch

import metrics_amd as ma
from metrics_amd.utilities.checks import check_forward_full_state_property


def test_compute_on_cpu_moves_list_states():
    m = ma.CatMetric(compute_on_cpu=True)
    m.update(torch.tensor([1.0, 2.0]))
    assert all(v.device.type == "cpu" for v in m.value)
    assert torch.equal(m.compute(), torch.tensor([1.0, 2.0]))


def test_set_dtype_real_metric():
    m = ma.MeanSquaredError()
    m.set_dtype(torch.float64)
    assert m.sum_squared_error.dtype == torch.float64
    m.update(torch.randn(8), torch.randn(8))
    assert m.compute().dtype == torch.float64


def test_check_forward_full_state_property_runs():
    check_forward_full_state_property(
        ma.MulticlassAccuracy,
        init_args={"num_classes": 3},
        input_args={"preds": torch.randn(16, 3), "target": torch.randint(0, 3, (16,))},
        num_update_to_compare=(2,),
        reps=1,
    )


def test_merge_state_real_metric():
    torch.manual_seed(0)
    preds = torch.randn(64, 5)
    target = torch.randint(0, 5, (64,))
    m1 = ma.MulticlassAccuracy(num_classes=5, average="micro")
    m2 = ma.MulticlassAccuracy(num_classes=5, average="micro")
    m1.update(preds[:32], target[:32])
    m2.update(preds[32:], target[32:])
    m1.merge_state(m2)
    ref = ma.MulticlassAccuracy(num_classes=5, average="micro")
    ref.update(preds, target)
    assert torch.allclose(m1.compute(), ref.compute())


def test_state_dict_roundtrip_collection():
    coll = ma.MetricCollection([ma.MulticlassAccuracy(num_classes=3), ma.MulticlassConfusionMatrix(num_classes=3)])
    coll.persistent(True)
    preds = torch.randn(32, 3)
    target = torch.randint(0, 3, (32,))
    coll.update(preds, target)
    sd = coll.state_dict()
    assert "MulticlassConfusionMatrix.confmat" in sd

    coll2 = ma.MetricCollection([ma.MulticlassAccuracy(num_classes=3), ma.MulticlassConfusionMatrix(num_classes=3)])
    coll2.persistent(True)
    coll2.load_state_dict(sd)
    r1 = coll.compute()
    r2 = coll2.compute()
    for k in r1:
        assert torch.allclose(r1[k].float(), r2[k].float())


def test_pickle_curve_metric_with_buffer():
    m = ma.BinaryAUROC(thresholds=50)
    m.update(torch.rand(64), torch.randint(0, 2, (64,)))
    m2 = pickle.loads(pickle.dumps(m))
    assert torch.allclose(m.compute(), m2.compute())


def test_forward_differentiable_metric():
    logits = torch.randn(32, requires_grad=True)
    target = torch.randint(0, 2, (32,))
    m = ma.BinaryHingeLoss()
    loss = m(logits, target)
    loss.backward()
    assert logits.grad is not None and logits.grad.abs().sum() > 0


def test_reset_after_sync_is_clean():
    m = ma.MulticlassAccuracy(num_classes=3, average="micro")
    m.update(torch.randn(16, 3), torch.randint(0, 3, (16,)))
    _ = m.compute()
    m.reset()
    assert m._update_count == 0
    assert (m.tp == 0).all()


def test_metric_in_module_device_move():
    class Model(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.acc = ma.MulticlassAccuracy(num_classes=3, average="micro")

    model = Model()
    model.to("cpu")
    assert model.acc.device == torch.device("cpu")
    sd = model.state_dict()  # metric states not persistent by default
    assert not any("acc" in k for k in sd)


def test_clone_with_collection_groups():
    coll = ma.MetricCollection([
        ma.MulticlassPrecision(num_classes=3, average="macro"),
        ma.MulticlassRecall(num_classes=3, average="macro"),
    ])
    preds = torch.randn(32, 3)
    target = torch.randint(0, 3, (32,))
    coll.update(preds, target)
    c2 = coll.clone(prefix="v_")
    r = c2.compute()
    assert set(r) == {"v_MulticlassPrecision", "v_MulticlassRecall"}


def test_clone_result_no_alias():
    """Batched result packing must not alias the inputs and must keep structure."""
    from metrics_amd.metric import _clone_result

    a = torch.arange(6.0)
    b = torch.ones(2, 3)
    c = torch.tensor(5.0)
    out = _clone_result({"x": a, "y": [b, c], "z": (a,)})
    assert torch.equal(out["x"], a) and out["x"].data_ptr() != a.data_ptr()
    assert torch.equal(out["y"][0], b) and out["y"][0].data_ptr() != b.data_ptr()
    assert torch.equal(out["y"][1], c)
    out["x"] += 1  # mutating the result must not touch the original
    assert torch.equal(a, torch.arange(6.0))


def test_compute_result_does_not_alias_state():
    import metrics_amd as ma

    m = ma.MulticlassConfusionMatrix(num_classes=3)
    m.update(torch.tensor([0, 1, 2]), torch.tensor([0, 1, 1]))
    res = m.compute()
    res += 100
    m._computed = None
    assert m.compute().max() < 100
