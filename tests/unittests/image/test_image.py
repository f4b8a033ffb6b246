"""Image metric tests (known values + invariants)."""
import math

import pytest
import torch

import metrics_amd as ma
from tests.unittests._helpers import seed_all


@pytest.fixture()
def imgs():
    seed_all(31)
    p = torch.rand(2, 3, 64, 64)
    t = torch.rand(2, 3, 64, 64)
    return p, t


def test_psnr_known(imgs):
    p, t = imgs
    v = ma.PeakSignalNoiseRatio(data_range=1.0)(p, t).item()
    mse = ((p - t) ** 2).mean().item()
    assert abs(v - 10 * math.log10(1.0 / mse)) < 1e-4


def test_psnr_accumulation(imgs):
    p, t = imgs
    m = ma.PeakSignalNoiseRatio(data_range=1.0)
    m.update(p[:1], t[:1])
    m.update(p[1:], t[1:])
    mse = ((p - t) ** 2).mean().item()
    assert abs(m.compute().item() - 10 * math.log10(1.0 / mse)) < 1e-4


def test_psnr_auto_range(imgs):
    p, t = imgs
    m = ma.PeakSignalNoiseRatio()
    m.update(p, t)
    rng = (t.max() - t.min()).item()
    mse = ((p - t) ** 2).mean().item()
    assert abs(m.compute().item() - 10 * math.log10(rng**2 / mse)) < 1e-3


def test_ssim_bounds(imgs):
    p, t = imgs
    assert abs(ma.StructuralSimilarityIndexMeasure(data_range=1.0)(p, p).item() - 1.0) < 1e-6
    v = ma.StructuralSimilarityIndexMeasure(data_range=1.0)(p, t).item()
    assert -1.0 <= v < 0.9


def test_ssim_shift_sensitivity():
    seed_all(32)
    base = torch.rand(1, 1, 64, 64)
    near = (base + 0.01 * torch.randn_like(base)).clamp(0, 1)
    far = (base + 0.3 * torch.randn_like(base)).clamp(0, 1)
    s_near = ma.StructuralSimilarityIndexMeasure(data_range=1.0)(near, base).item()
    s_far = ma.StructuralSimilarityIndexMeasure(data_range=1.0)(far, base).item()
    assert s_near > s_far


def test_ms_ssim(imgs):
    seed_all(33)
    p = torch.rand(1, 1, 256, 256)
    t = (p + 0.05 * torch.randn_like(p)).clamp(0, 1)
    v = ma.MultiScaleStructuralSimilarityIndexMeasure(data_range=1.0)(p, t).item()
    assert 0.5 < v <= 1.0
    assert abs(ma.MultiScaleStructuralSimilarityIndexMeasure(data_range=1.0)(p, p).item() - 1.0) < 1e-5


def test_uqi_sam_ergas_identity(imgs):
    p, _ = imgs
    assert abs(ma.UniversalImageQualityIndex()(p, p).item() - 1.0) < 1e-4
    assert ma.SpectralAngleMapper()(p, p).item() < 1e-3
    assert ma.ErrorRelativeGlobalDimensionlessSynthesis()(p, p).item() < 1e-3
    assert ma.RelativeAverageSpectralError()(p, p).item() < 1e-3


def test_total_variation_known():
    img = torch.zeros(1, 1, 3, 3)
    img[0, 0, 1, 1] = 1.0
    assert abs(ma.TotalVariation()(img).item() - 4.0) < 1e-6
    assert abs(ma.TotalVariation(reduction="mean")(img).item() - 4.0) < 1e-6


def test_rmse_sw_zero_and_positive(imgs):
    p, t = imgs
    assert ma.RootMeanSquaredErrorUsingSlidingWindow()(p, p).item() < 1e-6
    assert ma.RootMeanSquaredErrorUsingSlidingWindow()(p, t).item() > 0.1


def test_scc_identity(imgs):
    p, _ = imgs
    assert abs(ma.SpatialCorrelationCoefficient()(p, p).item() - 1.0) < 1e-4


def test_vif_identity():
    seed_all(34)
    x = torch.rand(1, 1, 64, 64)
    assert abs(ma.VisualInformationFidelity()(x, x).item() - 1.0) < 1e-4


def test_psnrb(imgs):
    p, t = imgs
    v = ma.PeakSignalNoiseRatioWithBlockedEffect()(p[:, :1], t[:, :1]).item()
    assert 0 < v < 20


def test_pansharpening_metrics():
    seed_all(35)
    fused = torch.rand(2, 4, 32, 32)  # high-res fused output
    ms = torch.rand(2, 4, 16, 16)  # low-res multispectral input
    pan = torch.rand(2, 4, 32, 32)  # high-res pan, one band per channel
    d_lambda = ma.SpectralDistortionIndex()(fused, ms).item()
    assert 0 <= d_lambda <= 1
    # identical fused == ms -> D_lambda == 0
    assert ma.SpectralDistortionIndex()(ms, ms).item() < 1e-6
    d_s = ma.SpatialDistortionIndex()(fused, {"ms": ms, "pan": pan}).item()
    assert 0 <= d_s <= 1
    qnr = ma.QualityWithNoReference()(fused, {"ms": ms, "pan": pan}).item()
    assert 0 <= qnr <= 1
    assert abs(qnr - (1 - d_lambda) * (1 - d_s)) < 1e-5
    # explicit pan_lr path
    pan_lr = torch.rand(2, 4, 16, 16)
    d_s2 = ma.SpatialDistortionIndex()(fused, {"ms": ms, "pan": pan, "pan_lr": pan_lr}).item()
    assert 0 <= d_s2 <= 1 and abs(d_s2 - d_s) > 1e-6


class _ToyFeat(torch.nn.Module):
    def __init__(self, d=24):
        super().__init__()
        torch.manual_seed(0)
        self.lin = torch.nn.Linear(3 * 16 * 16, d)

    def forward(self, x):
        return self.lin(x.float().flatten(1))


def test_fid_behaviour():
    seed_all(36)
    real = torch.rand(96, 3, 16, 16)
    close = (real + 0.01 * torch.randn_like(real)).clamp(0, 1)
    far = torch.rand(96, 3, 16, 16) * 0.3

    def fid_of(fake):
        m = ma.FrechetInceptionDistance(feature_extractor=_ToyFeat(), input_img_size=(3, 16, 16), normalize=True)
        m.update(real, real=True)
        m.update(fake, real=False)
        return m.compute().item()

    assert fid_of(close) < fid_of(far)


def test_fid_reset_real_features():
    real = torch.rand(8, 3, 16, 16)
    m = ma.FrechetInceptionDistance(
        feature_extractor=_ToyFeat(), input_img_size=(3, 16, 16), normalize=True, reset_real_features=False
    )
    m.update(real, real=True)
    n = m.real_features_num_samples.clone()
    m.reset()
    assert m.real_features_num_samples == n


def test_kid_and_is():
    seed_all(37)
    real = torch.rand(64, 3, 16, 16)
    fake = torch.rand(64, 3, 16, 16) * 0.5
    kid = ma.KernelInceptionDistance(feature_extractor=_ToyFeat(), subsets=3, subset_size=32)
    kid.update(real, real=True)
    kid.update(fake, real=False)
    km, ks = kid.compute()
    assert km.item() > 0
    isc = ma.InceptionScore(feature_extractor=_ToyFeat(), splits=2)
    isc.update(real)
    m, s = isc.compute()
    assert m.item() >= 1.0


def test_mifid():
    seed_all(38)
    real = torch.rand(64, 3, 16, 16)
    fake = torch.rand(64, 3, 16, 16)
    m = ma.MemorizationInformedFrechetInceptionDistance(feature_extractor=_ToyFeat())
    m.update(real, real=True)
    m.update(fake, real=False)
    assert m.compute().item() >= 0


def test_model_backed_raise_without_net():
    with pytest.raises(ModuleNotFoundError):
        ma.FrechetInceptionDistance()
    with pytest.raises(ModuleNotFoundError):
        ma.LearnedPerceptualImagePatchSimilarity()
    with pytest.raises(ModuleNotFoundError):
        ma.multimodal.CLIPScore()


def test_uniform_filter_matches_scipy():
    from scipy.ndimage import uniform_filter
    import numpy as np
    from metrics_amd.functional.image.misc import _scipy_uniform_filter

    seed_all(39)
    for w in (3, 7, 8):
        x = torch.rand(2, 3, 20, 24, dtype=torch.float64)
        mine = _scipy_uniform_filter(x, w)
        ref = torch.from_numpy(
            np.stack([[uniform_filter(x[b, c].numpy(), size=w, mode="reflect") for c in range(3)] for b in range(2)])
        )
        assert torch.allclose(mine, ref, atol=1e-12), w


def test_perceptual_path_length_toy():
    class ToyGen(torch.nn.Module):
        num_samples = None

        def __init__(self):
            super().__init__()
            torch.manual_seed(0)
            self.lin = torch.nn.Linear(8, 3 * 8 * 8)

        def sample(self, n):
            return torch.randn(n, 8)

        def forward(self, z):
            return self.lin(z).reshape(-1, 3, 8, 8)

    def sim(a, b):
        return ((a - b) ** 2).flatten(1).mean(1)

    m = ma.PerceptualPathLength(num_samples=64, batch_size=32, sim_net=sim, lower_discard=None, upper_discard=None)
    m.update(ToyGen())
    mean, std, med = m.compute()
    assert mean.item() > 0 and std.item() >= 0


def test_clip_iqa_and_infolm_raise_without_models():
    with pytest.raises(ModuleNotFoundError):
        ma.multimodal.CLIPImageQualityAssessment()
    with pytest.raises(ModuleNotFoundError):
        ma.text.InfoLM()


def _ref_image_functional(name):
    """Import a reference functional (offline oracle); None if unavailable."""
    import os
    import sys

    if not os.path.isdir("/root/reference/src"):
        return None
    sys.path.insert(0, "/root/repo/tools/refbench")
    sys.path.insert(0, "/root/reference/src")
    try:
        import torchmetrics.functional.image as ref_img
    except Exception:
        return None
    return getattr(ref_img, name)


@pytest.mark.parametrize(
    "kwargs",
    [
        {},
        {"gaussian_kernel": False, "kernel_size": 5},
        {"sigma": 1.0},
        {"return_contrast_sensitivity": True},
        {"reduction": "none"},
        {"data_range": (0.1, 0.9)},
    ],
)
def test_ssim_3d_vs_reference(kwargs):
    ref = _ref_image_functional("structural_similarity_index_measure")
    if ref is None:
        pytest.skip("reference tree not available")
    from metrics_amd.functional.image import structural_similarity_index_measure as ours

    torch.manual_seed(0)
    p = torch.rand(2, 1, 16, 16, 16)
    t = torch.rand(2, 1, 16, 16, 16)
    a, b = ours(p, t, **kwargs), ref(p, t, **kwargs)
    if isinstance(a, tuple):
        for x, y in zip(a, b):
            assert torch.allclose(x, y, atol=1e-6)
    else:
        assert torch.allclose(a, b, atol=1e-6)


def test_ms_ssim_3d_vs_reference():
    ref = _ref_image_functional("multiscale_structural_similarity_index_measure")
    if ref is None:
        pytest.skip("reference tree not available")
    from metrics_amd.functional.image import multiscale_structural_similarity_index_measure as ours

    torch.manual_seed(1)
    p = torch.rand(1, 1, 180, 180, 180)
    t = torch.rand(1, 1, 180, 180, 180)
    assert torch.allclose(ours(p, t), ref(p, t), atol=1e-6)


def test_ssim_3d_identity_and_bounds():
    """Self-contained (no reference needed): identical volumes score 1."""
    from metrics_amd.functional.image import structural_similarity_index_measure as ssim

    torch.manual_seed(2)
    p = torch.rand(2, 2, 12, 12, 12)
    assert torch.allclose(ssim(p, p.clone(), data_range=1.0), torch.tensor(1.0), atol=1e-5)
    t = torch.rand(2, 2, 12, 12, 12)
    v = ssim(p, t, data_range=1.0)
    assert -1.0 <= float(v) <= 1.0


def test_clip_score_math_with_toy_model():
    """CLIPScore math (normalized feature cosine * 100, clamp-at-0 mean)
    verified with a deterministic toy model + processor."""
    import torch

    import metrics_amd as ma

    class ToyModel:
        def eval(self):
            return self

        def get_image_features(self, pixel_values):
            return pixel_values.reshape(pixel_values.shape[0], -1)[:, :4].float()

        def get_text_features(self, input_ids, attention_mask):
            return input_ids[:, :4].float()

    class ToyProcessor:
        def __call__(self, text, images, return_tensors, padding):
            pix = torch.stack([i.reshape(-1)[:12].reshape(3, 2, 2).float() for i in images])
            ids = torch.stack([torch.arange(1, 5) * (len(t) % 7 + 1) for t in text])
            return {"pixel_values": pix, "input_ids": ids, "attention_mask": torch.ones_like(ids)}

    m = ma.multimodal.CLIPScore(model=ToyModel(), processor=ToyProcessor())
    g = torch.Generator().manual_seed(3)
    imgs = [torch.randint(0, 255, (3, 8, 8), generator=g).byte() for _ in range(3)]
    texts = ["a cat", "a dog on grass", "tree"]
    m.update(imgs, texts)
    out = m.compute()

    # hand-computed expectation
    proc = ToyProcessor()(texts, imgs, "pt", True)
    fi = ToyModel().get_image_features(proc["pixel_values"])
    fi = fi / fi.norm(p=2, dim=-1, keepdim=True)
    ft = ToyModel().get_text_features(proc["input_ids"], proc["attention_mask"])
    ft = ft / ft.norm(p=2, dim=-1, keepdim=True)
    exp = torch.clamp((100 * (fi * ft).sum(-1)).mean(), min=0)
    assert torch.allclose(out, exp, atol=1e-5)
