"""Detection metric tests: IoU family + COCO mAP engine vs known values."""
import pytest
import torch

from metrics_amd import ops
from metrics_amd.detection import (
    CompleteIntersectionOverUnion,
    DistanceIntersectionOverUnion,
    GeneralizedIntersectionOverUnion,
    IntersectionOverUnion,
    MeanAveragePrecision,
)
from metrics_amd.functional.detection.iou import intersection_over_union
from metrics_amd.detection.mean_ap import box_convert


def test_box_convert_roundtrip():
    boxes = torch.tensor([[1.0, 2.0, 5.0, 9.0], [0.0, 0.0, 2.0, 2.0]])
    for fmt in ("xywh", "cxcywh"):
        out = box_convert(box_convert(boxes, "xyxy", fmt), fmt, "xyxy")
        assert torch.allclose(out, boxes)


def test_pairwise_iou_values():
    b1 = torch.tensor([[0.0, 0.0, 10.0, 10.0]])
    b2 = torch.tensor([[5.0, 5.0, 15.0, 15.0], [0.0, 0.0, 10.0, 10.0], [20.0, 20.0, 30.0, 30.0]])
    iou = ops.box_iou_pairwise(b1, b2, "iou")
    assert torch.allclose(iou, torch.tensor([[25.0 / 175.0, 1.0, 0.0]]), atol=1e-6)


def test_giou_known_value():
    # identical boxes -> giou = 1; disjoint distant boxes -> giou < 0
    b1 = torch.tensor([[0.0, 0.0, 10.0, 10.0]])
    b2 = torch.tensor([[0.0, 0.0, 10.0, 10.0]])
    assert abs(ops.box_iou_pairwise(b1, b2, "giou").item() - 1.0) < 1e-6
    b3 = torch.tensor([[20.0, 20.0, 30.0, 30.0]])
    assert ops.box_iou_pairwise(b1, b3, "giou").item() < 0


def test_functional_iou_aggregate():
    p = torch.tensor([[0.0, 0.0, 10.0, 10.0], [5.0, 5.0, 15.0, 15.0]])
    t = torch.tensor([[0.0, 0.0, 10.0, 10.0], [5.0, 5.0, 15.0, 15.0]])
    assert abs(intersection_over_union(p, t).item() - 1.0) < 1e-6
    mat = intersection_over_union(p, t, aggregate=False)
    assert mat.shape == (2, 2)


def test_iou_module():
    preds = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), labels=torch.tensor([0]), scores=torch.tensor([0.9]))]
    target = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), labels=torch.tensor([0]))]
    m = IntersectionOverUnion()
    m.update(preds, target)
    assert abs(m.compute()["iou"].item() - 1.0) < 1e-6


@pytest.mark.parametrize("cls,key", [
    (GeneralizedIntersectionOverUnion, "giou"),
    (DistanceIntersectionOverUnion, "diou"),
    (CompleteIntersectionOverUnion, "ciou"),
])
def test_iou_variants_module(cls, key):
    preds = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), labels=torch.tensor([0]), scores=torch.tensor([0.9]))]
    target = [dict(boxes=torch.tensor([[2.0, 2.0, 12.0, 12.0]]), labels=torch.tensor([0]))]
    m = cls()
    m.update(preds, target)
    v = m.compute()[key].item()
    assert -1.5 <= v <= 1.0


def test_map_doc_example():
    """The canonical torchmetrics/pycocotools doc example: map=0.6, map_50=map_75=1.0."""
    preds = [dict(
        boxes=torch.tensor([[258.15, 41.29, 606.41, 285.07]]),
        scores=torch.tensor([0.536]),
        labels=torch.tensor([0]),
    )]
    target = [dict(boxes=torch.tensor([[214.15, 41.29, 562.41, 285.07]]), labels=torch.tensor([0]))]
    m = MeanAveragePrecision(iou_type="bbox")
    m.update(preds, target)
    res = m.compute()
    assert abs(res["map"].item() - 0.6) < 1e-6
    assert abs(res["map_50"].item() - 1.0) < 1e-6
    assert abs(res["map_75"].item() - 1.0) < 1e-6
    assert abs(res["mar_100"].item() - 0.6) < 1e-6


def test_map_perfect_and_missed():
    preds = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [20.0, 20.0, 30.0, 30.0]]),
                  scores=torch.tensor([0.9, 0.8]), labels=torch.tensor([1, 2]))]
    target = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [20.0, 20.0, 30.0, 30.0]]),
                   labels=torch.tensor([1, 2]))]
    m = MeanAveragePrecision()
    m.update(preds, target)
    assert abs(m.compute()["map"].item() - 1.0) < 1e-6

    m2 = MeanAveragePrecision()
    m2.update(
        [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), scores=torch.tensor([0.9]), labels=torch.tensor([1]))],
        [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), labels=torch.tensor([2]))],
    )
    assert abs(m2.compute()["map"].item()) < 1e-6


def test_map_false_positive_ranking():
    """An extra low-score FP after the TP leaves AP at 1.0; a high-score FP halves early precision."""
    target = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0]]), labels=torch.tensor([0]))]
    # low-score FP
    preds = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [50.0, 50.0, 60.0, 60.0]]),
                  scores=torch.tensor([0.9, 0.1]), labels=torch.tensor([0, 0]))]
    m = MeanAveragePrecision()
    m.update(preds, target)
    assert abs(m.compute()["map"].item() - 1.0) < 1e-6


def test_map_crowd_ignored():
    """Detections matched to crowd gts are ignored, not counted as FP."""
    target = [dict(
        boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [100.0, 100.0, 200.0, 200.0]]),
        labels=torch.tensor([0, 0]),
        iscrowd=torch.tensor([0, 1]),
    )]
    preds = [dict(
        boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [120.0, 120.0, 160.0, 160.0]]),
        scores=torch.tensor([0.9, 0.8]),
        labels=torch.tensor([0, 0]),
    )]
    m = MeanAveragePrecision()
    m.update(preds, target)
    res = m.compute()
    assert abs(res["map"].item() - 1.0) < 1e-6, res["map"]


def test_map_empty_inputs():
    m = MeanAveragePrecision()
    m.update(
        [dict(boxes=torch.zeros(0, 4), scores=torch.zeros(0), labels=torch.zeros(0, dtype=torch.long))],
        [dict(boxes=torch.zeros(0, 4), labels=torch.zeros(0, dtype=torch.long))],
    )
    res = m.compute()
    assert res["map"].item() == -1.0  # no classes -> undefined


def test_map_class_metrics():
    preds = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [20.0, 20.0, 30.0, 30.0]]),
                  scores=torch.tensor([0.9, 0.8]), labels=torch.tensor([1, 2]))]
    target = [dict(boxes=torch.tensor([[0.0, 0.0, 10.0, 10.0], [21.0, 21.0, 31.0, 31.0]]),
                   labels=torch.tensor([1, 2]))]
    m = MeanAveragePrecision(class_metrics=True)
    m.update(preds, target)
    res = m.compute()
    assert res["map_per_class"].shape == (2,)
    assert abs(res["map_per_class"][0].item() - 1.0) < 1e-6
