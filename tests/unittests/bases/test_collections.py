"""MetricCollection behavior tests (compute groups, renaming, nesting)."""
import pytest
import torch

import metrics_amd as ma
from metrics_amd import MetricCollection


def _inputs():
    torch.manual_seed(1)
    return torch.randn(64, 5), torch.randint(0, 5, (64,))


def _base_metrics():
    return [
        ma.MulticlassAccuracy(num_classes=5, average="macro"),
        ma.MulticlassPrecision(num_classes=5, average="macro"),
        ma.MulticlassRecall(num_classes=5, average="macro"),
        ma.MulticlassConfusionMatrix(num_classes=5),
    ]


def test_compute_group_formation():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics())
    coll.update(preds, target)
    groups = coll.compute_groups
    # acc/prec/recall share tp/fp/tn/fn -> one group; confmat its own
    flat = sorted(tuple(sorted(g)) for g in groups.values())
    assert ["MulticlassAccuracy", "MulticlassPrecision", "MulticlassRecall"] in [list(g) for g in flat]


def test_compute_groups_share_state_by_reference():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics())
    coll.update(preds, target)
    coll.update(preds, target)
    acc = coll._modules["MulticlassAccuracy"]
    prec = coll._modules["MulticlassPrecision"]
    assert acc.tp is prec.tp  # aliased


def test_items_access_breaks_aliasing():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics())
    coll.update(preds, target)
    items = dict(coll.items())  # copy_state=True default
    acc = items["MulticlassAccuracy"]
    prec = items["MulticlassPrecision"]
    assert acc.tp is not prec.tp
    assert torch.equal(acc.tp, prec.tp)
    # next update re-establishes links and stays correct
    coll.update(preds, target)
    res = coll.compute()
    single = ma.MulticlassAccuracy(num_classes=5, average="macro")
    single.update(preds, target)
    single.update(preds, target)
    assert torch.allclose(res["MulticlassAccuracy"], single.compute())


def test_compute_groups_results_match_disabled():
    preds, target = _inputs()
    coll_on = MetricCollection(_base_metrics(), compute_groups=True)
    coll_off = MetricCollection(_base_metrics(), compute_groups=False)
    for _ in range(3):
        coll_on.update(preds, target)
        coll_off.update(preds, target)
    r_on, r_off = coll_on.compute(), coll_off.compute()
    for k in r_on:
        assert torch.allclose(r_on[k].float(), r_off[k].float())


def test_user_compute_groups():
    coll = MetricCollection(
        _base_metrics()[:2],
        compute_groups=[["MulticlassAccuracy", "MulticlassPrecision"]],
    )
    preds, target = _inputs()
    coll.update(preds, target)
    assert coll.compute_groups == {0: ["MulticlassAccuracy", "MulticlassPrecision"]}


def test_prefix_postfix():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics()[:2], prefix="val_", postfix="_x")
    coll.update(preds, target)
    res = coll.compute()
    assert set(res) == {"val_MulticlassAccuracy_x", "val_MulticlassPrecision_x"}
    assert "val_MulticlassAccuracy_x" in coll.keys()


def test_clone_with_new_prefix():
    coll = MetricCollection(_base_metrics()[:1])
    c2 = coll.clone(prefix="train_")
    assert list(c2.keys()) == ["train_MulticlassAccuracy"]


def test_dict_input_sorted():
    coll = MetricCollection({
        "b_metric": ma.MulticlassAccuracy(num_classes=5),
        "a_metric": ma.MulticlassRecall(num_classes=5),
    })
    assert list(coll.keys(keep_base=True)) == ["a_metric", "b_metric"]


def test_nested_collections_flattened():
    inner = MetricCollection([ma.MulticlassAccuracy(num_classes=5)], prefix="in_")
    outer = MetricCollection({"grp": inner})
    preds, target = _inputs()
    outer.update(preds, target)
    res = outer.compute()
    assert any("in_" in k for k in res), res


def test_duplicate_names_raise():
    with pytest.raises(ValueError, match="two metrics both named"):
        MetricCollection([ma.MulticlassAccuracy(num_classes=5), ma.MulticlassAccuracy(num_classes=5)])


def test_invalid_input_raises():
    with pytest.raises(ValueError, match="Unknown input"):
        MetricCollection(42)


def test_forward_returns_dict_of_batch_values():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics()[:2])
    res = coll(preds, target)
    assert set(res) == {"MulticlassAccuracy", "MulticlassPrecision"}
    single = ma.MulticlassAccuracy(num_classes=5, average="macro")
    assert torch.allclose(res["MulticlassAccuracy"], single(preds, target))


def test_reset_all():
    preds, target = _inputs()
    coll = MetricCollection(_base_metrics())
    coll.update(preds, target)
    coll.reset()
    for m in coll.values(copy_state=False):
        assert m._update_count == 0


def test_kwarg_routing():
    """Metrics with different update signatures can coexist via kwarg filtering."""
    from metrics_amd import MeanMetric

    coll = MetricCollection({
        "acc": ma.MulticlassAccuracy(num_classes=5),
    })
    preds, target = _inputs()
    coll.update(preds=preds, target=target)
    assert coll.compute()["acc"] >= 0
