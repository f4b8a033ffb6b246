"""PanopticQuality / ModifiedPanopticQuality vs reference doctest values."""
import torch

import metrics_amd as ma


def test_panoptic_quality_reference_example():
    preds = torch.tensor([[[[6, 0], [0, 0], [6, 0], [6, 0]],
                           [[0, 0], [0, 0], [6, 0], [0, 1]],
                           [[0, 0], [0, 0], [6, 0], [0, 1]],
                           [[0, 0], [7, 0], [6, 0], [1, 0]],
                           [[0, 0], [7, 0], [7, 0], [7, 0]]]])
    target = torch.tensor([[[[6, 0], [0, 1], [6, 0], [0, 1]],
                            [[0, 1], [0, 1], [6, 0], [0, 1]],
                            [[0, 1], [0, 1], [6, 0], [1, 0]],
                            [[0, 1], [7, 0], [1, 0], [1, 0]],
                            [[0, 1], [7, 0], [7, 0], [7, 0]]]])
    pq = ma.detection.PanopticQuality(things={0, 1}, stuffs={6, 7})
    assert abs(float(pq(preds, target)) - 0.5463) < 1e-4


def test_modified_panoptic_quality_reference_example():
    preds = torch.tensor([[[0, 0], [0, 1], [6, 0], [7, 0], [0, 2], [1, 0]]])
    target = torch.tensor([[[0, 1], [0, 0], [6, 0], [7, 0], [6, 0], [255, 0]]])
    mpq = ma.detection.ModifiedPanopticQuality(things={0, 1}, stuffs={6, 7})
    assert abs(float(mpq(preds, target)) - 0.7667) < 1e-4


def test_panoptic_accumulation_and_reset():
    preds = torch.tensor([[[[6, 0], [0, 0]], [[7, 0], [1, 0]]]])
    target = preds.clone()
    pq = ma.detection.PanopticQuality(things={0, 1}, stuffs={6, 7})
    pq.update(preds, target)
    pq.update(preds, target)
    assert abs(float(pq.compute()) - 1.0) < 1e-6
    pq.reset()
    pq.update(preds, target)
    assert abs(float(pq.compute()) - 1.0) < 1e-6
