"""Task-dispatch wrappers + fairness/dice/nominal-matrix functionals."""
import pytest
import torch

from metrics_amd.functional import (
    cramers_v,
    cramers_v_matrix,
    demographic_parity,
    dice,
    equal_opportunity,
    image_gradients,
    logauc,
    pearsons_contingency_coefficient_matrix,
    precision_at_fixed_recall,
    recall_at_fixed_precision,
    sensitivity_at_specificity,
    specificity_at_sensitivity,
    theils_u,
    theils_u_matrix,
    tschuprows_t_matrix,
)
from metrics_amd.functional.classification import (
    binary_logauc,
    binary_recall_at_fixed_precision,
    multiclass_sensitivity_at_specificity,
)
from tests.unittests._helpers import seed_all


@pytest.fixture()
def binary_data():
    seed_all(41)
    return torch.rand(512), torch.randint(0, 2, (512,))


def test_task_wrappers_match_binary(binary_data):
    p, t = binary_data
    r1 = recall_at_fixed_precision(p, t, task="binary", min_precision=0.5)
    r2 = binary_recall_at_fixed_precision(p, t, min_precision=0.5)
    assert torch.equal(r1[0], r2[0]) and torch.equal(r1[1], r2[1])
    l1 = logauc(p, t, task="binary")
    l2 = binary_logauc(p, t)
    assert torch.equal(l1, l2)
    s1 = sensitivity_at_specificity(p, t, task="binary", min_specificity=0.6)
    assert 0 <= s1[0].item() <= 1
    s2 = specificity_at_sensitivity(p, t, task="binary", min_sensitivity=0.6)
    assert 0 <= s2[0].item() <= 1
    pr = precision_at_fixed_recall(p, t, task="binary", min_recall=0.5)
    assert 0 <= pr[0].item() <= 1


def test_task_wrappers_multiclass():
    seed_all(42)
    p = torch.randn(256, 5).softmax(-1)
    t = torch.randint(0, 5, (256,))
    r = recall_at_fixed_precision(p, t, task="multiclass", min_precision=0.3, num_classes=5)
    assert r[0].shape == (5,)
    s = sensitivity_at_specificity(p, t, task="multiclass", min_specificity=0.5, num_classes=5)
    assert torch.equal(s[0], multiclass_sensitivity_at_specificity(p, t, 5, 0.5)[0])
    v = logauc(p, t, task="multiclass", num_classes=5, average="macro")
    assert v.ndim == 0
    with pytest.raises(ValueError):
        recall_at_fixed_precision(p, t, task="multiclass", min_precision=0.3)  # num_classes missing


def test_fairness_wrappers():
    seed_all(43)
    preds = torch.rand(400)
    target = torch.randint(0, 2, (400,))
    groups = torch.randint(0, 3, (400,))
    dp = demographic_parity(preds, groups)
    assert len(dp) == 1 and all(k.startswith("DP_") for k in dp)
    assert 0 <= next(iter(dp.values())).item() <= 1
    eo = equal_opportunity(preds, target, groups)
    assert len(eo) == 1 and all(k.startswith("EO_") for k in eo)


def test_dice_functional_matches_modular():
    import metrics_amd as ma

    seed_all(44)
    p = torch.randint(0, 6, (300,))
    t = torch.randint(0, 6, (300,))
    for avg in ("micro", "macro"):
        f = dice(p, t, num_classes=6, average=avg)
        m = ma.Dice(num_classes=6, average=avg)
        m.update(p, t)
        assert torch.allclose(f.float(), m.compute().float())


def test_nominal_matrix_functions():
    seed_all(45)
    matrix = torch.randint(0, 4, (200, 4))
    cvm = cramers_v_matrix(matrix)
    assert cvm.shape == (4, 4)
    assert torch.allclose(cvm.diag(), torch.ones(4))
    assert torch.allclose(cvm, cvm.T)
    assert torch.allclose(cvm[0, 1], cramers_v(matrix[:, 0], matrix[:, 1]))
    pcm = pearsons_contingency_coefficient_matrix(matrix)
    assert torch.allclose(pcm, pcm.T)
    ttm = tschuprows_t_matrix(matrix)
    assert torch.allclose(ttm, ttm.T)
    tum = theils_u_matrix(matrix)
    assert torch.allclose(tum[1, 0], theils_u(matrix[:, 1], matrix[:, 0]))
    # Theil's U is asymmetric in general
    assert tum.shape == (4, 4)


def test_image_gradients_known():
    image = torch.arange(0, 25, dtype=torch.float32).reshape(1, 1, 5, 5)
    dy, dx = image_gradients(image)
    assert dy.shape == image.shape and dx.shape == image.shape
    assert torch.all(dy[0, 0, :4] == 5.0) and torch.all(dy[0, 0, 4] == 0.0)
    assert torch.all(dx[0, 0, :, :4] == 1.0) and torch.all(dx[0, 0, :, 4] == 0.0)
    with pytest.raises(RuntimeError):
        image_gradients(torch.zeros(5, 5))


def test_functional_perceptual_raise_without_net():
    from metrics_amd.functional.image import learned_perceptual_image_patch_similarity

    with pytest.raises(ModuleNotFoundError):
        learned_perceptual_image_patch_similarity(torch.rand(1, 3, 8, 8), torch.rand(1, 3, 8, 8))


def test_functional_clip_raise_without_model():
    from metrics_amd.functional.multimodal import clip_score

    with pytest.raises(ModuleNotFoundError):
        clip_score(torch.rand(3, 8, 8), "a photo")


def test_dice_without_num_classes_micro():
    """Legacy Dice infers the class count for micro averaging (reference parity)."""
    import metrics_amd as ma

    preds = torch.tensor([2, 0, 2, 1])
    target = torch.tensor([1, 1, 2, 0])
    assert ma.Dice(average="micro")(preds, target).item() == pytest.approx(0.25)
    assert ma.Dice()(preds, target).item() == pytest.approx(0.25)
    # probs input path
    probs = torch.nn.functional.one_hot(preds, 3).float()
    assert ma.Dice()(probs, target).item() == pytest.approx(0.25)


def test_input_transformer_passes_kwargs_through():
    """BinaryTargetTransformer forwards retrieval-style kwargs untouched."""
    import metrics_amd as ma  # noqa: F401
    from metrics_amd.retrieval import RetrievalMRR
    from metrics_amd.wrappers import BinaryTargetTransformer

    torch.manual_seed(0)
    preds = torch.rand(10)
    topics = torch.randint(0, 2, (10,))
    targets = torch.randint(0, 5, (10,))
    m = BinaryTargetTransformer(RetrievalMRR(), threshold=2)
    m.update(preds, targets, indexes=topics)
    expected = RetrievalMRR()
    expected.update(preds, (targets > 2).long(), indexes=topics)
    assert torch.allclose(m.compute(), expected.compute())
