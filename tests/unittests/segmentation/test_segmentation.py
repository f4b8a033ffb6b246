"""Segmentation metric tests."""
import torch

from metrics_amd.segmentation import DiceScore, GeneralizedDiceScore, HausdorffDistance, MeanIoU


def _onehot(x, c):
    return torch.nn.functional.one_hot(x, num_classes=c).movedim(-1, 1)


def test_mean_iou_perfect():
    t = torch.randint(0, 3, (2, 16, 16))
    m = MeanIoU(num_classes=3, input_format="index")
    m.update(t, t)
    assert abs(m.compute().item() - 1.0) < 1e-6


def test_mean_iou_known():
    pred = torch.tensor([[0, 0, 1, 1]])
    tgt = torch.tensor([[0, 1, 1, 1]])
    m = MeanIoU(num_classes=2, input_format="index")
    m.update(pred, tgt)
    # class0: inter 1, union 2 -> 0.5 ; class1: inter 2, union 3 -> 2/3
    assert abs(m.compute().item() - (0.5 + 2 / 3) / 2) < 1e-6


def test_dice_micro():
    pred = torch.tensor([[0, 0, 1, 1]])
    tgt = torch.tensor([[0, 1, 1, 1]])
    m = DiceScore(num_classes=2, average="micro", input_format="index")
    m.update(pred, tgt)
    # micro: num = 2*(1+2)=6, denom = 4+4=8
    assert abs(m.compute().item() - 6 / 8) < 1e-6


def test_generalized_dice_perfect():
    t = torch.randint(0, 3, (2, 8, 8))
    m = GeneralizedDiceScore(num_classes=3, input_format="index")
    m.update(t, t)
    assert abs(m.compute().item() - 1.0) < 1e-5


def test_hausdorff_zero_for_identical():
    t = torch.zeros(1, 16, 16, dtype=torch.long)
    t[0, 4:10, 4:10] = 1
    m = HausdorffDistance(num_classes=2, input_format="index")
    m.update(t, t)
    assert m.compute().item() == 0.0


def test_hausdorff_known_shift():
    a = torch.zeros(1, 16, 16, dtype=torch.long)
    a[0, 2:6, 2:6] = 1
    b = torch.zeros(1, 16, 16, dtype=torch.long)
    b[0, 2:6, 5:9] = 1  # shifted 3 in x
    m = HausdorffDistance(num_classes=2, input_format="index")
    m.update(a, b)
    assert abs(m.compute().item() - 3.0) < 1e-6
