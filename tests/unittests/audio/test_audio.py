"""Audio metric tests."""
import torch

import metrics_amd as ma
from tests.unittests._helpers import seed_all
from metrics_amd.functional.audio import (
    permutation_invariant_training,
    scale_invariant_signal_noise_ratio,
    signal_distortion_ratio,
)


def _sig(n=8000, b=2, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(b, n, generator=g)


def test_snr_known():
    t = _sig()
    noise = 0.1 * _sig(seed=1)
    p = t + noise
    ref = (10 * torch.log10((t**2).sum(-1) / (noise**2).sum(-1))).mean().item()
    assert abs(ma.SignalNoiseRatio()(p, t).item() - ref) < 1e-4


def test_si_snr_scale_invariance():
    t = _sig()
    p = t + 0.1 * _sig(seed=1)
    v1 = ma.ScaleInvariantSignalNoiseRatio()(p, t).item()
    v2 = ma.ScaleInvariantSignalNoiseRatio()(p, 5.0 * t).item()
    assert abs(v1 - v2) < 1e-3


def test_si_sdr_known():
    # torchmetrics doc example
    target = torch.tensor([3.0, -0.5, 2.0, 7.0])
    preds = torch.tensor([2.5, 0.0, 2.0, 8.0])
    v = ma.ScaleInvariantSignalDistortionRatio()(preds, target).item()
    assert abs(v - 18.4030) < 1e-3, v


def test_sdr_identical_high():
    t = _sig()
    assert signal_distortion_ratio(t, t).mean().item() > 40


def test_sdr_batch_accumulate():
    t = _sig(b=4)
    p = t + 0.05 * _sig(b=4, seed=2)
    m = ma.SignalDistortionRatio()
    m.update(p[:2], t[:2])
    m.update(p[2:], t[2:])
    v = m.compute().item()
    ref = signal_distortion_ratio(p, t).mean().item()
    assert abs(v - ref) < 1e-3


def test_sa_sdr():
    t = torch.randn(2, 3, 4000)
    v = ma.SourceAggregatedSignalDistortionRatio()(t, t).item()
    assert v > 40


def test_complex_si_snr():
    spec = torch.randn(1, 129, 20, 2)
    v = ma.ComplexScaleInvariantSignalNoiseRatio()(spec, spec).item()
    assert v > 40


def test_pit_recovers_permutation():
    s = torch.randn(3, 2, 4000)
    shuffled = s[:, [1, 0], :]
    best, perm = permutation_invariant_training(shuffled, s, scale_invariant_signal_noise_ratio, eval_func="max")
    assert (perm == torch.tensor([1, 0])).all()
    m = ma.PermutationInvariantTraining(scale_invariant_signal_noise_ratio)
    m.update(shuffled, s)
    assert m.compute().item() > 40


def test_pit_permutation_wise_matches_speaker_wise():
    s = torch.randn(2, 3, 1000)
    p = s + 0.1 * torch.randn_like(s)
    b1, _ = permutation_invariant_training(p, s, scale_invariant_signal_noise_ratio, mode="speaker-wise")
    b2, _ = permutation_invariant_training(p, s, scale_invariant_signal_noise_ratio, mode="permutation-wise")
    assert torch.allclose(b1, b2, atol=1e-4)


def test_external_dsp_metrics_raise():
    import pytest

    with pytest.raises(ModuleNotFoundError):
        ma.audio.PerceptualEvaluationSpeechQuality(16000, "wb")
    with pytest.raises(ModuleNotFoundError):
        ma.audio.ShortTimeObjectiveIntelligibility(16000)


def test_sdr_cg_matches_direct_solve():
    from metrics_amd.functional.audio import signal_distortion_ratio

    seed_all(61)
    target = torch.randn(2, 8000)
    preds = target + 0.1 * torch.randn(2, 8000)
    direct = signal_distortion_ratio(preds, target)
    cg = signal_distortion_ratio(preds, target, use_cg_iter=20)
    assert torch.allclose(direct, cg, atol=0.05), (direct, cg)


def test_external_dsp_stubs_raise():
    """Reference behavior when optional DSP deps are missing: constructor raises."""
    import pytest as _pytest

    for cls, kwargs in [
        (ma.audio.DeepNoiseSuppressionMeanOpinionScore, {"fs": 16000, "personalized": False}),
        (ma.audio.NonIntrusiveSpeechQualityAssessment, {"fs": 16000}),
        (ma.audio.SpeechReverberationModulationEnergyRatio, {"fs": 16000}),
        (ma.audio.PerceptualEvaluationSpeechQuality, {"fs": 16000, "mode": "wb"}),
        (ma.audio.ShortTimeObjectiveIntelligibility, {"fs": 16000}),
    ]:
        with _pytest.raises(ModuleNotFoundError):
            cls(**kwargs)
