"""Audio metrics that wrap external DSP packages (pesq, pystoi, gammatone,
onnxruntime/librosa) — none of which are installable in this offline image.

Parity: reference ``functional/audio/{pesq,stoi,srmr,dnsmos,nisqa}.py``. Like
the reference when its optional dependency is absent, each function raises
``ModuleNotFoundError`` naming the package to install.
"""
from __future__ import annotations

from typing import Optional

from torch import Tensor


def _require(pkg: str, extra: str) -> None:
    raise ModuleNotFoundError(
        f"Metric requires that `{pkg}` is installed. Either install as `pip install metrics-amd[{extra}]`"
        f" or `pip install {pkg}`."
    )


def perceptual_evaluation_speech_quality(
    preds: Tensor,
    target: Tensor,
    fs: int,
    mode: str,
    keep_same_device: bool = False,
    n_processes: int = 1,
) -> Tensor:
    """PESQ via the `pesq` package (reference functional/audio/pesq.py:26)."""
    _require("pesq", "audio")


def short_time_objective_intelligibility(
    preds: Tensor,
    target: Tensor,
    fs: int,
    extended: bool = False,
    keep_same_device: bool = False,
) -> Tensor:
    """STOI via the `pystoi` package (reference functional/audio/stoi.py:25)."""
    _require("pystoi", "audio")


def speech_reverberation_modulation_energy_ratio(
    preds: Tensor,
    fs: int,
    n_cochlear_filters: int = 23,
    low_freq: float = 125,
    min_cf: float = 4,
    max_cf: Optional[float] = None,
    norm: bool = False,
    fast: bool = False,
) -> Tensor:
    """SRMR via `gammatone`+`torchaudio` (reference functional/audio/srmr.py:176)."""
    _require("gammatone", "audio")


def deep_noise_suppression_mean_opinion_score(
    preds: Tensor,
    fs: int,
    personalized: bool,
    device: Optional[str] = None,
    num_threads: Optional[int] = None,
    cache_session: bool = True,
) -> Tensor:
    """DNSMOS via `librosa`+`onnxruntime` (reference functional/audio/dnsmos.py:182)."""
    _require("librosa", "audio")


def non_intrusive_speech_quality_assessment(preds: Tensor, fs: int) -> Tensor:
    """NISQA via `librosa`+`requests` (reference functional/audio/nisqa.py:66)."""
    _require("librosa", "audio")


__all__ = [
    "deep_noise_suppression_mean_opinion_score",
    "non_intrusive_speech_quality_assessment",
    "perceptual_evaluation_speech_quality",
    "short_time_objective_intelligibility",
    "speech_reverberation_modulation_energy_ratio",
]
