"""Audio metrics (functional): SNR / SI-SNR / SDR / SI-SDR / SA-SDR / complex SI-SNR / PIT.

Parity: torchmetrics ``functional/audio/{snr,sdr,pit}.py``. SDR follows the
fast_bss_eval formulation: FFT autocorrelation + symmetric-Toeplitz system for
the optimal distortion filter (rocFFT via torch.fft; dense solve via rocSOLVER).
"""
from __future__ import annotations

from typing import Any, Callable, Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.utilities.checks import _check_same_shape


def signal_noise_ratio(preds: Tensor, target: Tensor, zero_mean: bool = False) -> Tensor:
    """SNR = 10 log10(||target||^2 / ||target - preds||^2)."""
    _check_same_shape(preds, target)
    eps = torch.finfo(preds.dtype).eps

    if zero_mean:
        target = target - torch.mean(target, dim=-1, keepdim=True)
        preds = preds - torch.mean(preds, dim=-1, keepdim=True)

    noise = target - preds

    snr_value = (torch.sum(target**2, dim=-1) + eps) / (torch.sum(noise**2, dim=-1) + eps)
    return 10 * torch.log10(snr_value)


def scale_invariant_signal_noise_ratio(preds: Tensor, target: Tensor) -> Tensor:
    """SI-SNR (always zero-mean + optimal scaling of the target)."""
    return scale_invariant_signal_distortion_ratio(preds, target, zero_mean=True)


def scale_invariant_signal_distortion_ratio(preds: Tensor, target: Tensor, zero_mean: bool = False) -> Tensor:
    """SI-SDR."""
    _check_same_shape(preds, target)
    eps = torch.finfo(preds.dtype).eps

    if zero_mean:
        target = target - torch.mean(target, dim=-1, keepdim=True)
        preds = preds - torch.mean(preds, dim=-1, keepdim=True)

    alpha = (torch.sum(preds * target, dim=-1, keepdim=True) + eps) / (
        torch.sum(target**2, dim=-1, keepdim=True) + eps
    )
    target_scaled = alpha * target

    noise = target_scaled - preds

    val = (torch.sum(target_scaled**2, dim=-1) + eps) / (torch.sum(noise**2, dim=-1) + eps)
    return 10 * torch.log10(val)


def source_aggregated_signal_distortion_ratio(
    preds: Tensor, target: Tensor, scale_invariant: bool = True, zero_mean: bool = False
) -> Tensor:
    """SA-SDR: one SDR over the aggregate of all sources (preds/target (..., spk, time))."""
    _check_same_shape(preds, target)
    if preds.ndim < 2:
        raise RuntimeError(f"Expected inputs with at least 2 dims (..., spk, time), got {preds.shape}")
    eps = torch.finfo(preds.dtype).eps
    if zero_mean:
        target = target - torch.mean(target, dim=-1, keepdim=True)
        preds = preds - torch.mean(preds, dim=-1, keepdim=True)
    if scale_invariant:
        # one alpha shared by all speakers (shape [..., 1, 1]) — that is the
        # "source-aggregated" part of SA-SDR
        alpha = ((preds * target).sum(dim=(-2, -1), keepdim=True) + eps) / (
            (target**2).sum(dim=(-2, -1), keepdim=True) + eps
        )
        target = alpha * target
    distortion = target - preds
    val = ((target**2).sum(dim=(-2, -1)) + eps) / ((distortion**2).sum(dim=(-2, -1)) + eps)
    return 10 * torch.log10(val)


def complex_scale_invariant_signal_noise_ratio(preds: Tensor, target: Tensor, zero_mean: bool = False) -> Tensor:
    """C-SI-SNR on complex-valued spectrograms (..., frequency, time, 2) or complex dtype."""
    if preds.is_complex():
        preds = torch.view_as_real(preds)
        target = torch.view_as_real(target)
    if preds.shape != target.shape or preds.shape[-1] != 2:
        raise RuntimeError(
            "Predictions and targets are expected to have the shape (..., frequency, time, 2), but got"
            f" {preds.shape} and {target.shape}."
        )
    # flatten complex spectrogram into a real vector per sample
    preds = preds.reshape(*preds.shape[:-3], -1)
    target = target.reshape(*target.shape[:-3], -1)
    return scale_invariant_signal_distortion_ratio(preds, target, zero_mean=zero_mean)


def _symmetric_toeplitz(vector: Tensor) -> Tensor:
    """Construct a symmetric Toeplitz matrix from its first row (batched)."""
    vec_exp = torch.cat([vector.flip(-1), vector[..., 1:]], dim=-1)
    v_len = vector.shape[-1]
    return torch.as_strided(
        vec_exp, size=(*vector.shape[:-1], v_len, v_len), stride=(*vec_exp.stride()[:-1], 1, 1)
    ).flip(dims=(-1,))


def signal_distortion_ratio(
    preds: Tensor,
    target: Tensor,
    use_cg_iter: Optional[int] = None,
    filter_length: int = 512,
    zero_mean: bool = False,
    load_diag: Optional[float] = None,
) -> Tensor:
    """SDR with an optimal length-``filter_length`` distortion filter (fast_bss_eval formulation)."""
    _check_same_shape(preds, target)

    # use double precision
    preds_dtype = preds.dtype
    preds = preds.double()
    target = target.double()

    if zero_mean:
        preds = preds - preds.mean(dim=-1, keepdim=True)
        target = target - target.mean(dim=-1, keepdim=True)

    # normalize along time-axis to make the bss_eval equations well-conditioned
    preds = preds / (preds.norm(dim=-1, keepdim=True) + 1e-38)
    target = target / (target.norm(dim=-1, keepdim=True) + 1e-38)

    length = target.shape[-1] + filter_length - 1
    n_fft = 2 ** int(torch.ceil(torch.log2(torch.tensor(length, dtype=torch.float))).item())

    t_fft = torch.fft.rfft(target, n=n_fft, dim=-1)
    p_fft = torch.fft.rfft(preds, n=n_fft, dim=-1)

    # auto-correlation of target (first filter_length lags)
    acf = torch.fft.irfft(t_fft.abs() ** 2, n=n_fft, dim=-1)[..., :filter_length]
    # cross-correlation target x preds
    xcorr = torch.fft.irfft(t_fft.conj() * p_fft, n=n_fft, dim=-1)[..., :filter_length]

    if load_diag is not None:
        acf = acf.clone()
        acf[..., 0] += load_diag

    r = _symmetric_toeplitz(acf)
    if use_cg_iter is not None:
        # batched conjugate gradient (fast_bss_eval's use_cg_iter option): the
        # normalized system is well-conditioned, so a few matvecs beat the
        # O(L^3) dense solve; each matvec is one batched O(L^2) matmul
        b = xcorr
        sol = torch.zeros_like(b)
        resid = b.clone()
        p = resid.clone()
        rs_old = (resid * resid).sum(dim=-1, keepdim=True)
        for _ in range(int(use_cg_iter)):
            ap = (r @ p.unsqueeze(-1)).squeeze(-1)
            alpha = rs_old / ((p * ap).sum(dim=-1, keepdim=True) + 1e-38)
            sol = sol + alpha * p
            resid = resid - alpha * ap
            rs_new = (resid * resid).sum(dim=-1, keepdim=True)
            p = resid + (rs_new / rs_old) * p
            rs_old = rs_new
    else:
        sol = torch.linalg.solve(r, xcorr.unsqueeze(-1)).squeeze(-1)

    # coherence: <xcorr, sol>
    coh = (xcorr * sol).sum(dim=-1)
    ratio = coh / (1 - coh + 1e-38)
    val = 10 * torch.log10(ratio.clamp(min=1e-38))
    return val.to(preds_dtype)


def permutation_invariant_training(
    preds: Tensor,
    target: Tensor,
    metric_func: Callable,
    mode: str = "speaker-wise",
    eval_func: str = "max",
    **kwargs: Any,
) -> Tuple[Tensor, Tensor]:
    """PIT: best metric over source permutations; returns (best values, permutations)."""
    if preds.shape[0:2] != target.shape[0:2]:
        raise RuntimeError(
            "Predictions and targets are expected to have the same shape at the batch and speaker dimensions"
        )
    if eval_func not in ("max", "min"):
        raise ValueError(f'eval_func can only be "max" or "min" but got {eval_func}')
    if mode not in ("speaker-wise", "permutation-wise"):
        raise ValueError(f'mode can only be "speaker-wise" or "permutation-wise" but got {mode}')
    if target.ndim < 2:
        raise ValueError(f"Inputs must be of shape [batch, spk, ...], got {target.shape} and {preds.shape} instead")

    batch_size, spk_num = target.shape[0:2]

    if mode == "speaker-wise":
        # pairwise metric matrix (batch, preds_spk, target_spk)
        metric_mtx = torch.empty(batch_size, spk_num, spk_num, device=target.device)
        for t in range(spk_num):
            for e in range(spk_num):
                metric_mtx[:, t, e] = metric_func(preds[:, e, ...], target[:, t, ...], **kwargs)
        # Hungarian assignment per batch item (scipy on CPU; tiny matrices)
        from scipy.optimize import linear_sum_assignment

        mm = metric_mtx.detach().cpu().numpy()
        best_metric = torch.empty(batch_size, device=target.device, dtype=metric_mtx.dtype)
        best_perm = torch.empty(batch_size, spk_num, device=target.device, dtype=torch.long)
        for b in range(batch_size):
            row, col = linear_sum_assignment(mm[b], maximize=eval_func == "max")
            best_metric[b] = metric_mtx[b, row, col].mean()
            perm = torch.empty(spk_num, dtype=torch.long)
            perm[torch.from_numpy(row)] = torch.from_numpy(col)
            best_perm[b] = perm
        return best_metric, best_perm

    # permutation-wise: enumerate all permutations
    import itertools

    # the metric sees whole (batch, spk, ...) tensors with preds permuted, all
    # permutations batched into one call (reference functional/audio/pit.py:173-188);
    # per-speaker metric outputs are averaged over trailing dims
    perms = torch.tensor(list(itertools.permutations(range(spk_num))), device=preds.device)
    perm_num = perms.shape[0]
    ppreds = torch.index_select(preds, dim=1, index=perms.reshape(-1)).reshape(
        batch_size * perm_num, *preds.shape[1:]
    )
    ptarget = target.repeat_interleave(repeats=perm_num, dim=0)
    vals = metric_func(ppreds, ptarget, **kwargs)
    vals_t = torch.mean(vals.reshape(batch_size, perm_num, -1), dim=-1)  # (batch, n_perms)
    if eval_func == "max":
        best, idx = vals_t.max(dim=1)
    else:
        best, idx = vals_t.min(dim=1)
    best_perm = perms[idx.detach(), :]
    return best, best_perm


def pit_permutate(preds: Tensor, perm: Tensor) -> Tensor:
    """Reorder sources by the permutation returned from PIT."""
    return torch.stack([torch.index_select(pred, 0, p) for pred, p in zip(preds, perm)])
