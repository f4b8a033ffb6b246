"""ctypes bindings for the in-tree gfx950 HIP kernel library.

Every wrapper takes torch CUDA (= HIP on ROCm) tensors, validates layout,
and launches on the *current* torch stream — so kernels order correctly with
surrounding torch ops without any global synchronization.

If the library is missing on a GPU machine, these raise loudly: metric
updates must never silently fall back to stock torch kernels on MI355X.
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional, Tuple

import torch
from torch import Tensor

from metrics_amd.csrc.build import lib_path

_LIB: Optional[ctypes.CDLL] = None
_LOAD_ERR: Optional[str] = None


def _load() -> Optional[ctypes.CDLL]:
    global _LIB, _LOAD_ERR
    if _LIB is not None:
        return _LIB
    p = lib_path()
    if not p.exists():
        _LOAD_ERR = f"HIP kernel library not built at {p}. Run `python -m metrics_amd.csrc.build`."
        return None
    try:
        _LIB = ctypes.CDLL(str(p))
    except OSError as err:
        _LOAD_ERR = f"Failed to load {p}: {err}"
        return None
    _declare_argtypes(_LIB)
    return _LIB


_U64 = ctypes.c_uint64
_LL = ctypes.c_longlong
_I = ctypes.c_int
_F = ctypes.c_float
_D = ctypes.c_double

_SIGNATURES = {
    "ma_mc_stat_logits": [_U64, _U64, _I, _U64, _LL, _LL, _LL, _I, _U64, _U64, _U64, _U64, _U64, _U64, _U64, _U64, _U64],
    "ma_mc_stat_labels": [_U64, _U64, _U64, _LL, _LL, _LL, _I, _U64, _U64, _U64, _U64, _U64],
    "ma_bincount": [_U64, _U64, _LL, _LL, _U64],
    "ma_binary_stat": [_U64, _U64, _I, _U64, _LL, _F, _LL, _I, _U64, _U64],
    "ma_multilabel_stat": [_U64, _U64, _I, _U64, _LL, _LL, _F, _LL, _I, _U64, _U64],
    "ma_binary_curve_hist": [_U64, _U64, _I, _U64, _LL, _U64, _I, _LL, _I, _I, _F, _F, _I, _U64, _U64],
    "ma_multiclass_curve_hist": [_U64, _U64, _I, _U64, _LL, _LL, _U64, _I, _LL, _I, _I, _I, _F, _F, _I, _U64, _U64, _U64, _I, _I, _I, _U64],
    "ma_curve_suffix": [_U64, _U64, _LL, _I, _I, _I, _U64, _U64],
    "ma_curve_epoch_bump": [_U64, _U64],
    "ma_confmat_scalars": [_U64, _U64, _LL, _U64, ctypes.c_float, _U64],
    "ma_linear_stat_multi": [_U64, _U64, _U64, _U64, _U64, _LL, _I, _U64, _U64],
    "ma_apply_stat_deltas": [_U64, _U64, _LL, _U64, _U64, _U64, _U64, _U64],
    "ma_exact_apply": [_U64, _U64, _LL, _LL, _I, _U64, _U64],
    "ma_apply_stat_exact": [_U64, _U64, _LL, _LL, _U64, _U64, _U64, _U64, _U64, _U64, _U64],
    "ma_curve_auc_from_confmat": [_U64, _U64, _I, _LL, _I, _U64, _U64],
    "ma_linear_stat_compute": [_U64, _U64, _U64, _U64, _U64, _LL, _F, _F, _F, _F, _F, _F, _F, _F, _I, _I, _F, _F, _F, _U64],
    "ma_err_reduce": [_U64, _U64, _U64, _I, _LL, _I, _D, _U64, _I, _I, _U64],
    "ma_box_iou": [_U64, _U64, _LL, _U64, _LL, _I, _U64],
    "ma_clf_curve_scratch_bytes": [_LL, _I, ctypes.POINTER(ctypes.c_ulonglong)],
    "ma_binary_clf_curve": [_U64, _U64, _U64, _U64, _LL, _LL, _U64, ctypes.c_ulonglong, _U64, _U64, _U64, _U64],
    "ma_ssim2d_fused": [_U64, _U64, _U64, _I, _LL, _LL, _LL, _LL, _U64, _I, _U64, _I, _F, _F, _U64, _F, _F, _I, _I, _I, _U64, _U64],
    "ma_binary_erosion2d": [_U64, _U64, _LL, _LL, _LL, _LL, _U64, _I, _I, _I, _I, _I, _U64],
    "ma_calib_bins": [_U64, _U64, _U64, _LL, _U64, _I, _I, _F, _F, _U64],
    "ma_mc_clf_curve_scratch_bytes": [_LL, _LL, ctypes.POINTER(ctypes.c_ulonglong)],
    "ma_mc_clf_curve": [_U64, _U64, _U64, _LL, _LL, _I, _U64, ctypes.c_ulonglong, _U64, _U64, _U64, _U64],
    "ma_retrieval_sort_scratch_bytes": [_LL, ctypes.POINTER(ctypes.c_ulonglong)],
    "ma_retrieval_sort": [_U64, _U64, _U64, _LL, _U64, ctypes.c_ulonglong, _U64, _U64],
    "ma_mc_topk_stat": [_U64, _U64, _I, _U64, _LL, _LL, _I, _LL, _I, _U64, _U64, _U64, _U64],
}


def _declare_argtypes(lib: ctypes.CDLL) -> None:
    for name, argtypes in _SIGNATURES.items():
        fn = getattr(lib, name)
        fn.argtypes = argtypes
        fn.restype = ctypes.c_int


def hip_available() -> bool:
    """True if the kernel library is built and loadable."""
    return _load() is not None


def _lib() -> ctypes.CDLL:
    lib = _load()
    if lib is None:
        raise RuntimeError(
            f"metrics_amd HIP extension required for GPU tensors but unavailable: {_LOAD_ERR}"
        )
    return lib


try:
    # raw-stream fast path (same API torch.compile generated code uses):
    # torch.cuda.current_stream() builds a Stream object + device-index
    # lookups per call (~9us measured) — pure dispatch overhead at our call
    # rates (every kernel launch goes through _stream()).
    _raw_stream = torch._C._cuda_getCurrentRawStream
except AttributeError:  # pragma: no cover - older torch
    _raw_stream = None


def _stream() -> int:
    if _raw_stream is not None:
        return _raw_stream(torch.cuda.current_device())
    return torch.cuda.current_stream().cuda_stream


def _check(rc: int, name: str) -> None:
    if rc != 0:
        raise RuntimeError(f"HIP kernel {name} failed with error code {rc}")


def _dtype_code(t: Tensor) -> int:
    if t.dtype == torch.float32:
        return 0
    if t.dtype == torch.bfloat16:
        return 1
    raise TypeError(f"HIP metric kernels support float32/bfloat16 inputs, got {t.dtype}")


def _to_supported(t: Tensor) -> Tensor:
    """Kernels read fp32/bf16; cast anything else (fp16 under AMP, fp64) to fp32."""
    if t.is_floating_point() and t.dtype not in (torch.float32, torch.bfloat16):
        return t.float()
    return t


def mc_stat_logits(
    preds: Tensor, target: Tensor, ignore_index: Optional[int], want_confmat: bool,
    want_argmax: bool = False,
) -> Tuple[Tensor, Tensor, Tensor, Tensor, Optional[Tensor], Optional[Tensor]]:
    """Fused row-argmax + per-class tp/fp/fn (+confmat) over (B, C) logits.

    Returns (tp, fp, fn, valid_count, confmat|None, argmax|None), all int64 on device.
    """
    lib = _lib()
    assert preds.ndim == 2 and target.ndim == 1 and preds.shape[0] == target.shape[0]
    preds = _to_supported(preds).contiguous()
    target = target.contiguous().long()
    B, C = preds.shape
    dev = preds.device
    tp = torch.zeros(C, dtype=torch.long, device=dev)
    fp = torch.zeros(C, dtype=torch.long, device=dev)
    fn = torch.zeros(C, dtype=torch.long, device=dev)
    valid = torch.zeros(1, dtype=torch.long, device=dev)
    confmat = torch.zeros(C, C, dtype=torch.long, device=dev) if want_confmat else None
    argmax = torch.empty(B, dtype=torch.long, device=dev) if want_argmax else None
    rc = lib.ma_mc_stat_logits(
        _stream(),
        preds.data_ptr(),
        _dtype_code(preds),
        target.data_ptr(),
        B,
        C,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        tp.data_ptr(),
        fp.data_ptr(),
        fn.data_ptr(),
        confmat.data_ptr() if confmat is not None else 0,
        valid.data_ptr(),
        argmax.data_ptr() if argmax is not None else 0,
        0, 0, 0,
    )
    _check(rc, "ma_mc_stat_logits")
    return tp, fp, fn, valid, confmat, argmax


def mc_stat_labels(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int], want_confmat: bool
) -> Tuple[Tensor, Tensor, Tensor, Tensor, Optional[Tensor]]:
    """Per-class tp/fp/fn (+confmat) from integer label predictions."""
    lib = _lib()
    preds = preds.contiguous().long().flatten()
    target = target.contiguous().long().flatten()
    N = preds.numel()
    dev = preds.device
    tp = torch.zeros(num_classes, dtype=torch.long, device=dev)
    fp = torch.zeros(num_classes, dtype=torch.long, device=dev)
    fn = torch.zeros(num_classes, dtype=torch.long, device=dev)
    valid = torch.zeros(1, dtype=torch.long, device=dev)
    confmat = torch.zeros(num_classes, num_classes, dtype=torch.long, device=dev) if want_confmat else None
    rc = lib.ma_mc_stat_labels(
        _stream(),
        preds.data_ptr(),
        target.data_ptr(),
        N,
        num_classes,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        tp.data_ptr(),
        fp.data_ptr(),
        fn.data_ptr(),
        confmat.data_ptr() if confmat is not None else 0,
        valid.data_ptr(),
    )
    _check(rc, "ma_mc_stat_labels")
    return tp, fp, fn, valid, confmat


def bincount(x: Tensor, minlength: int) -> Tensor:
    """Deterministic LDS-privatized histogram of a non-negative int64 tensor."""
    lib = _lib()
    x = x.contiguous().long().flatten()
    out = torch.zeros(minlength, dtype=torch.long, device=x.device)
    rc = lib.ma_bincount(
        _stream(),
        x.data_ptr(),
        x.numel(),
        minlength,
        out.data_ptr(),
    )
    _check(rc, "ma_bincount")
    return out


def binary_stat(
    preds: Tensor, target: Tensor, threshold: float, ignore_index: Optional[int]
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Fused binary tp/fp/tn/fn in one pass; picks raw vs sigmoid thresholding on-device."""
    lib = _lib()
    preds = _to_supported(preds).contiguous()
    target = target.contiguous().long()
    N = preds.numel()
    dev = preds.device
    out = torch.zeros(2, 4, dtype=torch.long, device=dev)  # {raw,sig} x {tp,fp,tn,fn}
    flag = torch.zeros(1, dtype=torch.int32, device=dev)
    # no clamp: at thr=0/1 the kernel's logit(thr) is -inf/+inf, which makes the
    # sigmoid branch always/never positive — exactly the reference comparison
    thr = float(threshold)
    rc = lib.ma_binary_stat(
        _stream(),
        preds.data_ptr(),
        _dtype_code(preds),
        target.data_ptr(),
        N,
        thr,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        out.data_ptr(),
        flag.data_ptr(),
    )
    _check(rc, "ma_binary_stat")
    sel = flag.long().squeeze(0)  # 0 -> raw counts, 1 -> sigmoid counts (no host sync)
    counts = out[0] * (1 - sel) + out[1] * sel
    return counts[0], counts[1], counts[2], counts[3]


def multilabel_stat(
    preds: Tensor, target: Tensor, threshold: float, ignore_index: Optional[int]
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Fused per-label tp/fp/tn/fn over (N, L)."""
    lib = _lib()
    assert preds.ndim == 2
    preds = _to_supported(preds).contiguous()
    target = target.contiguous().long()
    N, L = preds.shape
    dev = preds.device
    out = torch.zeros(2, L, 4, dtype=torch.long, device=dev)
    flag = torch.zeros(1, dtype=torch.int32, device=dev)
    thr = float(threshold)
    rc = lib.ma_multilabel_stat(
        _stream(),
        preds.data_ptr(),
        _dtype_code(preds),
        target.data_ptr(),
        N,
        L,
        thr,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        out.data_ptr(),
        flag.data_ptr(),
    )
    _check(rc, "ma_multilabel_stat")
    sel = flag.long().squeeze(0)
    counts = out[0] * (1 - sel) + out[1] * sel  # (L, 4)
    return counts[:, 0], counts[:, 1], counts[:, 2], counts[:, 3]


_UNIFORM_CACHE: dict = {}

# Per-device DEVICE-side epoch buffer E (2 x uint32): E[1] counts completed
# curve updates (bumped by the suffix kernel), E[0] records the epoch whose
# inputs were out-of-range. "Normalize this update" <=> E[0] == E[1]+1. All
# state lives on device, so the whole protocol is hipGraph-capturable.
# Stream-safety: launches go to the caller's current stream; each
# (detect -> histogram -> suffix) triple is issued back-to-back on one stream.
_FLAG_BUFS: dict = {}
_HIST_POOL: dict = {}
_ROWSTATS_POOL: dict = {}
# -1 auto, 0 wave-aggregated global atomics, 1 LDS-privatized (A/B probe knob)
_CURVE_VARIANT = int(os.environ.get("METRICS_AMD_CURVE_VARIANT", "-1"))
_CURVE_CCHUNK = int(os.environ.get("METRICS_AMD_CURVE_CCHUNK", "0"))


def _epoch_buf(device, owner=None) -> Tensor:
    """Device-epoch buffer E (2 x uint32). Owned per-METRIC when ``owner`` is
    given so independent metrics can update on parallel streams without racing
    on the protocol state; falls back to a per-device buffer otherwise."""
    if owner is not None:
        buf = owner.__dict__.get("_hip_epoch_buf")
        if buf is None or buf.device != device:
            buf = torch.zeros(2, dtype=torch.int32, device=device)
            owner.__dict__["_hip_epoch_buf"] = buf
        return buf
    key = device.index
    buf = _FLAG_BUFS.get(key)
    if buf is None:
        buf = torch.zeros(2, dtype=torch.int32, device=device)
        _FLAG_BUFS[key] = buf
    return buf


def _pooled_hist(outer: int, T: int, device, owner=None) -> Tensor:
    """Persistent histogram scratch, zeroed once; the suffix kernel re-zeroes
    it in-flight after consuming it (zero_hist=1), so reuse needs no fill.
    Owned per-METRIC when ``owner`` is given (parallel-stream safety)."""
    if owner is not None:
        buf = owner.__dict__.get("_hip_hist_buf")
        if buf is None or buf.device != device or buf.shape != (outer, T + 1, 2):
            buf = torch.zeros(outer, T + 1, 2, dtype=torch.long, device=device)
            owner.__dict__["_hip_hist_buf"] = buf
        return buf
    key = (outer, T, device.index)
    buf = _HIST_POOL.get(key)
    if buf is None:
        buf = torch.zeros(outer, T + 1, 2, dtype=torch.long, device=device)
        if len(_HIST_POOL) > 64:
            _HIST_POOL.clear()
        _HIST_POOL[key] = buf
    return buf


def _uniform_params(thr: Tensor):
    """Detect uniform threshold spacing for the O(1) bucket guess.

    The kernel's +-1 fixup keeps results exact even if the guess is off.
    Result is cached per (data_ptr, numel): thresholds are fixed per metric,
    and the detection needs device->host syncs we don't want per update.
    """
    key = (thr.data_ptr(), thr.numel(), thr.device.index)
    hit = _UNIFORM_CACHE.get(key)
    if hit is not None:
        return hit
    T = thr.numel()
    if T < 2:
        out = (0, 0.0, 1.0)
    else:
        tc = thr.detach().float().cpu()
        t0, t_last = float(tc[0]), float(tc[-1])
        if t_last <= t0:
            out = (0, 0.0, 1.0)
        else:
            step = (t_last - t0) / (T - 1)
            diffs = tc[1:] - tc[:-1]
            # strictly uniform within small tolerance -> O(1) guess stays O(1)
            uniform = bool(((diffs - step).abs() <= 1e-3 * step + 1e-9).all())
            out = (1 if uniform else 0, t0, (T - 1) / (t_last - t0))
    if len(_UNIFORM_CACHE) > 256:
        _UNIFORM_CACHE.clear()
    _UNIFORM_CACHE[key] = out
    return out


def binary_curve_confmat(
    preds: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int]
) -> Tensor:
    """(T,2,2) threshold confmat via bucketized histogram + on-device suffix-sum.

    ``preds`` must already be probabilities in [0,1].
    """
    lib = _lib()
    preds = _to_supported(preds).contiguous()
    target = target.contiguous().long()
    thr = thresholds.contiguous().float()
    T = thr.numel()
    dev = preds.device
    uni, t0, inv_step = _uniform_params(thr)
    hist = torch.zeros(T + 1, 2, dtype=torch.long, device=dev)
    rc = lib.ma_binary_curve_hist(
        _stream(),
        preds.data_ptr(),
        _dtype_code(preds),
        target.data_ptr(),
        preds.numel(),
        thr.data_ptr(),
        T,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        uni,
        t0,
        inv_step,
        0,
        0,
        hist.data_ptr(),
    )
    _check(rc, "ma_binary_curve_hist")
    confmat = torch.zeros(T, 2, 2, dtype=torch.long, device=dev)
    rc = lib.ma_curve_suffix(
        _stream(),
        hist.data_ptr(),
        1,
        T,
        0,
        0,
        0,
        confmat.data_ptr(),
    )
    _check(rc, "ma_curve_suffix")
    return confmat


def multiclass_curve_confmat(
    probs: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int], mode: int = 0
) -> Tensor:
    """(T,C,2,2) threshold confmats. mode 0: multiclass one-vs-rest (target (B,));
    mode 1: multilabel (target (B,C)). ``probs`` already normalized (B,C)."""
    lib = _lib()
    probs = _to_supported(probs).contiguous()
    target = target.contiguous().long()
    thr = thresholds.contiguous().float()
    B, C = probs.shape
    T = thr.numel()
    dev = probs.device
    uni, t0, inv_step = _uniform_params(thr)
    hist = torch.zeros(C, T + 1, 2, dtype=torch.long, device=dev)
    rc = lib.ma_multiclass_curve_hist(
        _stream(),
        probs.data_ptr(),
        _dtype_code(probs),
        target.data_ptr(),
        B,
        C,
        thr.data_ptr(),
        T,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        mode,
        uni,
        t0,
        inv_step,
        0,
        0,
        0,
        0,
        _CURVE_VARIANT,
        _CURVE_CCHUNK,
        0,
        hist.data_ptr(),
    )
    _check(rc, "ma_multiclass_curve_hist")
    # write the (T, C, 2, 2) state layout directly (transposed suffix kernel)
    confmat = torch.zeros(T, C, 2, 2, dtype=torch.long, device=dev)
    rc = lib.ma_curve_suffix(
        _stream(),
        hist.data_ptr(),
        C,
        T,
        1,
        0,
        0,
        confmat.data_ptr(),
    )
    _check(rc, "ma_curve_suffix")
    return confmat


_CLF_SCRATCH_CACHE: dict = {}


def binary_clf_curve(
    preds: Tensor, target: Tensor, weights: Optional[Tensor] = None, pos_label: int = 1
) -> Tuple[Tensor, Tensor, Tensor]:
    """Exact (fps, tps, thresholds) at distinct descending scores.

    rocPRIM radix sort (stable, fp32 keys) + fp64 device scans + fused
    gather/flag and compaction kernels — the K2 path for thresholds=None
    ROC/PR/AUROC. One D2H sync for the data-dependent output length (the
    reference's torch path syncs there too).
    """
    lib = _lib()
    preds = _to_supported(preds)
    if preds.dtype != torch.float32:
        preds = preds.float()
    preds = preds.contiguous().flatten()
    target = target.contiguous().long().flatten()
    w = weights.contiguous().float().flatten() if weights is not None else None
    N = preds.numel()
    dev = preds.device
    key = (N, w is not None)
    nbytes = _CLF_SCRATCH_CACHE.get(key)
    if nbytes is None:
        out_b = ctypes.c_ulonglong(0)
        rc = lib.ma_clf_curve_scratch_bytes(N, 1 if w is not None else 0, ctypes.byref(out_b))
        _check(rc, "ma_clf_curve_scratch_bytes")
        nbytes = out_b.value
        if len(_CLF_SCRATCH_CACHE) > 256:
            _CLF_SCRATCH_CACHE.clear()
        _CLF_SCRATCH_CACHE[key] = nbytes
    scratch = torch.empty(int(nbytes), dtype=torch.uint8, device=dev)
    out_fps = torch.empty(N, dtype=torch.float32, device=dev)
    out_tps = torch.empty(N, dtype=torch.float32, device=dev)
    out_thr = torch.empty(N, dtype=torch.float32, device=dev)
    out_cnt = torch.zeros(1, dtype=torch.long, device=dev)
    rc = lib.ma_binary_clf_curve(
        _stream(),
        preds.data_ptr(),
        target.data_ptr(),
        w.data_ptr() if w is not None else 0,
        N,
        pos_label,
        scratch.data_ptr(),
        nbytes,
        out_fps.data_ptr(),
        out_tps.data_ptr(),
        out_thr.data_ptr(),
        out_cnt.data_ptr(),
    )
    _check(rc, "ma_binary_clf_curve")
    k = int(out_cnt.item())
    return out_fps[:k], out_tps[:k], out_thr[:k]


_MC_CURVE_SCRATCH_CACHE: dict = {}


def mc_clf_curve(probs: Tensor, target: Tensor, multilabel: bool = False):
    """Per-class exact curves for ALL classes in one composite-key sort.

    ``probs`` (B, C); ``target`` (B,) class ids (multiclass) or (B, C) 0/1
    (multilabel). Returns a list of C ``(fps, tps, thresholds)`` tuples —
    one device sort + ONE host count transfer instead of C sorts + C syncs.
    """
    lib = _lib()
    probs = _to_supported(probs)
    if probs.dtype != torch.float32:
        probs = probs.float()
    probs = probs.contiguous()
    target = target.contiguous().long()
    B, C = probs.shape
    n = B * C
    dev = probs.device
    nbytes = _MC_CURVE_SCRATCH_CACHE.get((B, C))
    if nbytes is None:
        out_b = ctypes.c_ulonglong(0)
        rc = lib.ma_mc_clf_curve_scratch_bytes(B, C, ctypes.byref(out_b))
        _check(rc, "ma_mc_clf_curve_scratch_bytes")
        nbytes = out_b.value
        if len(_MC_CURVE_SCRATCH_CACHE) > 256:
            _MC_CURVE_SCRATCH_CACHE.clear()
        _MC_CURVE_SCRATCH_CACHE[(B, C)] = nbytes
    scratch = torch.empty(int(nbytes), dtype=torch.uint8, device=dev)
    out_fps = torch.empty(n, dtype=torch.float32, device=dev)
    out_tps = torch.empty(n, dtype=torch.float32, device=dev)
    out_thr = torch.empty(n, dtype=torch.float32, device=dev)
    out_counts = torch.zeros(C, dtype=torch.long, device=dev)
    rc = lib.ma_mc_clf_curve(
        _stream(),
        probs.data_ptr(),
        target.data_ptr(),
        B,
        C,
        1 if multilabel else 0,
        scratch.data_ptr(),
        nbytes,
        out_fps.data_ptr(),
        out_tps.data_ptr(),
        out_thr.data_ptr(),
        out_counts.data_ptr(),
    )
    _check(rc, "ma_mc_clf_curve")
    # per-class positive counts ride the same single host transfer: the
    # callers' "no positive / no negative samples" branches then need no
    # further syncs (neg_c = B - pos_c)
    if multilabel:
        pos = target.sum(0)
    else:
        pos = torch.bincount(target.clamp(min=0), minlength=C)[:C]
    packed = torch.cat([out_counts, pos]).cpu()
    counts = packed[:C].tolist()
    pos_counts = packed[C:].tolist()
    res = []
    off = 0
    for c in range(C):
        k = counts[c]
        res.append(
            (out_fps[off : off + k], out_tps[off : off + k], out_thr[off : off + k],
             int(pos_counts[c]), B - int(pos_counts[c]))
        )
        off += k
    return res


_RETR_SCRATCH_CACHE: dict = {}


def retrieval_sort(indexes: Tensor, preds: Tensor) -> Tuple[Tensor, Tensor]:
    """(order, by_index) permutations for retrieval grouping in ONE radix pass.

    ``order`` sorts by (query index asc, pred desc); ``by_index`` by index
    only with original order preserved for ties. Composite 64-bit keys
    replace the torch double-argsort lexsort. Requires indexes < 2^31.
    """
    lib = _lib()
    indexes = indexes.contiguous().long()
    preds = _to_supported(preds).float().contiguous()
    n = preds.numel()
    dev = preds.device
    nbytes = _RETR_SCRATCH_CACHE.get(n)
    if nbytes is None:
        out_b = ctypes.c_ulonglong(0)
        rc = lib.ma_retrieval_sort_scratch_bytes(n, ctypes.byref(out_b))
        _check(rc, "ma_retrieval_sort_scratch_bytes")
        nbytes = out_b.value
        if len(_RETR_SCRATCH_CACHE) > 256:
            _RETR_SCRATCH_CACHE.clear()
        _RETR_SCRATCH_CACHE[n] = nbytes
    scratch = torch.empty(int(nbytes), dtype=torch.uint8, device=dev)
    order = torch.empty(n, dtype=torch.int32, device=dev)
    by_index = torch.empty(n, dtype=torch.int32, device=dev)
    rc = lib.ma_retrieval_sort(
        _stream(),
        indexes.data_ptr(),
        preds.data_ptr(),
        n,
        scratch.data_ptr(),
        nbytes,
        order.data_ptr(),
        by_index.data_ptr(),
    )
    _check(rc, "ma_retrieval_sort")
    return order.long(), by_index.long()


_ERR_OPS = {"sq_err": (0, 1), "abs_err": (1, 1), "ape": (2, 1), "sq_log_err": (3, 1), "moments": (4, 6), "logcosh": (5, 1)}


def err_reduce(x: Tensor, y: Tensor, op: str, eps: float = 1.17e-6) -> Tensor:
    """Deterministic fused elementwise-error reduction; returns fp64 sums (n_out,)."""
    lib = _lib()
    op_id, n_out = _ERR_OPS[op]
    x = x.contiguous()
    y = y.contiguous()
    N = x.numel()
    num_blocks = min(max((N + 255) // 256, 1), 2048)
    dev = x.device
    partials = torch.zeros(num_blocks, n_out, dtype=torch.float64, device=dev)
    out = torch.zeros(n_out, dtype=torch.float64, device=dev)
    rc = lib.ma_err_reduce(
        _stream(),
        x.data_ptr(),
        y.data_ptr(),
        _dtype_code(x),
        N,
        op_id,
        eps,
        partials.data_ptr(),
        num_blocks,
        n_out,
        out.data_ptr(),
    )
    _check(rc, "ma_err_reduce")
    return out


_CALIB_SCALE = 8589934592.0  # 2^33, matches csrc/kernels2.hip


def calib_bins(conf: Tensor, acc: Tensor, bounds: Tensor) -> Tuple[Tensor, Tensor, Tensor]:
    """Fused bucketize + per-bin (count, Σconf, Σacc) — K5 kernel.

    Integer fixed-point accumulation => deterministic. Returns fp32
    (acc_bin_sum, conf_bin_sum, count_bin) of length ``len(bounds)-1``.
    """
    lib = _lib()
    conf = _to_supported(conf).float().contiguous().flatten()
    acc = acc.float().contiguous().flatten()
    bounds = bounds.float().contiguous()
    n_bins = bounds.numel() - 1
    uni, b0, inv_step = _uniform_params(bounds)
    out = torch.zeros(3, n_bins, dtype=torch.long, device=conf.device)
    rc = lib.ma_calib_bins(
        _stream(),
        conf.data_ptr(),
        acc.data_ptr(),
        conf.numel(),
        bounds.data_ptr(),
        n_bins,
        uni,
        b0,
        inv_step,
        out.data_ptr(),
    )
    if rc == 9001:
        raise RuntimeError("calibration bin count exceeds the LDS budget")
    _check(rc, "ma_calib_bins")
    count = out[0].float()
    conf_sum = out[1].double().div_(_CALIB_SCALE).float()
    acc_sum = out[2].double().div_(_CALIB_SCALE).float()
    return acc_sum, conf_sum, count


def mc_topk_stat(
    preds: Tensor, target: Tensor, num_classes: int, k: int, ignore_index: Optional[int]
) -> Tuple[Tensor, Tensor, Tensor, Tensor]:
    """Per-class (tp, fp, tn, fn) for top-k multiclass stat scores — K3 kernel.

    torchmetrics semantics: the effective prediction is the target when it
    ranks in the top-k, else the argmax (_refine_preds_oh).
    """
    lib = _lib()
    assert preds.ndim == 2 and preds.is_floating_point()
    preds = _to_supported(preds).contiguous()
    target = target.contiguous().long()
    B, C = preds.shape
    dev = preds.device
    tp = torch.zeros(C, dtype=torch.long, device=dev)
    fp = torch.zeros(C, dtype=torch.long, device=dev)
    fn = torch.zeros(C, dtype=torch.long, device=dev)
    valid = torch.zeros(1, dtype=torch.long, device=dev)
    rc = lib.ma_mc_topk_stat(
        _stream(),
        preds.data_ptr(),
        _dtype_code(preds),
        target.data_ptr(),
        B,
        C,
        k,
        ignore_index if ignore_index is not None else 0,
        1 if ignore_index is not None else 0,
        tp.data_ptr(),
        fp.data_ptr(),
        fn.data_ptr(),
        valid.data_ptr(),
    )
    _check(rc, "ma_mc_topk_stat")
    tn = valid - (tp + fp + fn)
    return tp, fp, tn, fn


class SsimLdsOverflow(RuntimeError):
    """Window too large for the one-kernel LDS tiling; caller falls back."""


def ssim2d_fused(
    preds: Tensor,
    target: Tensor,
    wh: Tensor,
    ww: Tensor,
    c1: float,
    c2: float,
    dr: Optional[Tensor],
    k1: float,
    k2: float,
    want_cs: bool,
    crop_h: int,
    crop_w: int,
) -> Tuple[Tensor, Optional[Tensor]]:
    """Fused 2D SSIM: per-image Σssim (and optional cropped Σcs) in ONE kernel.

    ``wh``/``ww`` are the separable window weight vectors (odd length);
    ``dr`` an optional 0-dim float32 device tensor holding data_range (c1/c2
    then computed in-kernel — no host sync for the data_range=None path).
    """
    lib = _lib()
    preds = _to_supported(preds).contiguous()
    target = target.to(preds.dtype).contiguous()
    B, C, H, W = preds.shape
    rh = (wh.numel() - 1) // 2
    rw = (ww.numel() - 1) // 2
    dev = preds.device
    wh = wh.contiguous().float()
    ww = ww.contiguous().float()
    sum_sim = torch.zeros(B, dtype=torch.float64, device=dev)
    sum_cs = torch.zeros(B, dtype=torch.float64, device=dev) if want_cs else None
    rc = lib.ma_ssim2d_fused(
        _stream(),
        preds.data_ptr(),
        target.data_ptr(),
        _dtype_code(preds),
        B, C, H, W,
        wh.data_ptr(), rh,
        ww.data_ptr(), rw,
        c1, c2,
        dr.data_ptr() if dr is not None else 0,
        k1, k2,
        1 if want_cs else 0,
        crop_h, crop_w,
        sum_sim.data_ptr(),
        sum_cs.data_ptr() if sum_cs is not None else 0,
    )
    if rc == 9001:
        raise SsimLdsOverflow(f"window radius ({rh},{rw}) exceeds the LDS tile budget")
    _check(rc, "ma_ssim2d_fused")
    return sum_sim, sum_cs


def binary_erosion2d(img: Tensor, strel: Tensor, origin: Tuple[int, int], border_value: int) -> Tensor:
    """Windowed erosion of a (N,C,H,W) binary uint8 image; returns uint8."""
    lib = _lib()
    img = img.contiguous()
    assert img.dtype == torch.uint8 and img.ndim == 4
    N, C, H, W = img.shape
    strel = strel.contiguous().int()
    kh, kw = strel.shape
    out = torch.empty_like(img)
    rc = lib.ma_binary_erosion2d(
        _stream(),
        img.data_ptr(),
        N, C, H, W,
        strel.data_ptr(),
        kh, kw,
        origin[0], origin[1],
        int(border_value),
        out.data_ptr(),
    )
    _check(rc, "ma_binary_erosion2d")
    return out


def box_iou(boxes1: Tensor, boxes2: Tensor, variant: str = "iou") -> Tensor:
    """All-pairs (N,M) box IoU / GIoU / DIoU / CIoU on xyxy boxes."""
    lib = _lib()
    v = {"iou": 0, "giou": 1, "diou": 2, "ciou": 3}[variant]
    boxes1 = boxes1.contiguous().float()
    boxes2 = boxes2.contiguous().float()
    N, M = boxes1.shape[0], boxes2.shape[0]
    out = torch.empty(N, M, dtype=torch.float32, device=boxes1.device)
    if N == 0 or M == 0:
        return out
    rc = lib.ma_box_iou(
        _stream(),
        boxes1.data_ptr(),
        N,
        boxes2.data_ptr(),
        M,
        v,
        out.data_ptr(),
    )
    _check(rc, "ma_box_iou")
    return out


def curve_hist_into_confmat(
    probs: Tensor, target: Tensor, thresholds: Tensor, ignore_index: Optional[int],
    confmat_state: Tensor, mode: int, norm: Optional[str] = None, owner=None,
    stats_ready: bool = False, lazy: Optional[bool] = None,
    skip_epoch_bump: bool = False,
) -> None:
    """Bucketized histogram + transposed suffix-sum accumulated DIRECTLY into the
    metric's confmat state ((T,2,2) binary / (T,C,2,2) multiclass|multilabel) —
    no intermediate confmat materialization or host-side add.

    ``norm`` ("softmax" / "sigmoid" / None) moves the reference's
    normalize-iff-outside-[0,1] semantics INTO the kernels: a fused row-stats
    (or range-detect) pass sets a per-device epoch flag and the histogram
    kernel normalizes inline — replacing the 4-kernel torch chain
    (compare / any-reduce / softmax / where) and its 3 extra full-tensor
    round-trips through HBM. The histogram scratch is pooled and re-zeroed by
    the suffix kernel in-flight, so steady-state updates launch no fills.
    """
    lib = _lib()
    thr = thresholds.contiguous().float()
    T = thr.numel()
    uni, t0, inv_step = _uniform_params(thr)
    dev = probs.device
    if confmat_state.ndim == 3:  # binary (T,2,2)
        preds = _to_supported(probs).contiguous().flatten()
        tgt = target.contiguous().long().flatten()
        hist = _pooled_hist(1, T, dev, owner)
        if norm == "sigmoid":
            norm_i, flag_ptr = 1, _epoch_buf(dev, owner).data_ptr()
        else:
            norm_i, flag_ptr = 0, 0
        rc = lib.ma_binary_curve_hist(
            _stream(),
            preds.data_ptr(),
            _dtype_code(preds),
            tgt.data_ptr(),
            preds.numel(),
            thr.data_ptr(),
            T,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            uni,
            t0,
            inv_step,
            norm_i,
            flag_ptr,
            hist.data_ptr(),
        )
        _check(rc, "ma_binary_curve_hist")
        outer, transposed = 1, 0
    else:
        probs = _to_supported(probs).contiguous()
        tgt = target.contiguous().long()
        B, C = probs.shape
        hist = _pooled_hist(C, T, dev, owner)
        if norm == "softmax":
            if owner is not None:
                rbuf = owner.__dict__.get("_hip_rowstats_buf")
                if rbuf is None or rbuf.device != dev or rbuf.shape[1] < B:
                    rbuf = torch.empty(2, B, dtype=torch.float32, device=dev)
                    owner.__dict__["_hip_rowstats_buf"] = rbuf
            else:
                rkey = (B, dev.index)
                rbuf = _ROWSTATS_POOL.get(rkey)
                if rbuf is None:
                    rbuf = torch.empty(2, B, dtype=torch.float32, device=dev)
                    if len(_ROWSTATS_POOL) > 64:
                        _ROWSTATS_POOL.clear()
                    _ROWSTATS_POOL[rkey] = rbuf
            norm_i, flag_ptr = 1, _epoch_buf(dev, owner).data_ptr()
            rm_ptr, ri_ptr = rbuf[0].data_ptr(), rbuf[1].data_ptr()
        elif norm == "sigmoid":
            norm_i, flag_ptr, rm_ptr, ri_ptr = 2, _epoch_buf(dev, owner).data_ptr(), 0, 0
        else:
            norm_i, flag_ptr, rm_ptr, ri_ptr = 0, 0, 0, 0
        rc = lib.ma_multiclass_curve_hist(
            _stream(),
            probs.data_ptr(),
            _dtype_code(probs),
            tgt.data_ptr(),
            B,
            C,
            thr.data_ptr(),
            T,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            mode,
            uni,
            t0,
            inv_step,
            norm_i,
            flag_ptr,
            rm_ptr,
            ri_ptr,
            _CURVE_VARIANT,
            _CURVE_CCHUNK,
            1 if stats_ready else 0,
            hist.data_ptr(),
        )
        _check(rc, "ma_multiclass_curve_hist")
        outer, transposed = C, 1
    assert confmat_state.is_contiguous()
    # Lazy mode (default for metric-owned updates): histograms accumulate
    # additively across updates in the per-owner buffer; the O(T*outer)
    # suffix/confmat materialization is DEFERRED to the first state read
    # (compute/sync/state_dict/forward — Metric._maybe_flush_lazy). The
    # per-update device epoch is closed by a 1-thread bump kernel instead of
    # the suffix kernel. Cuts a ~22us/step kernel at the bench shape.
    if lazy is None:
        lazy = owner is not None
    if lazy and owner is not None:
        if norm is not None and not skip_epoch_bump:
            rc = lib.ma_curve_epoch_bump(_stream(), _epoch_buf(dev, owner).data_ptr())
            _check(rc, "ma_curve_epoch_bump")
        owner.__dict__["_hip_lazy_meta"] = (outer, T, transposed)
        owner.__dict__["_lazy_dirty"] = True
        return
    rc = lib.ma_curve_suffix(
        _stream(),
        hist.data_ptr(),
        outer,
        T,
        transposed,
        1,  # re-zero the pooled hist in-flight
        _epoch_buf(dev, owner).data_ptr() if norm is not None else 0,
        confmat_state.data_ptr(),
    )
    _check(rc, "ma_curve_suffix")


import weakref as _weakref

# Tensors cannot key a WeakKeyDictionary (their __eq__ is elementwise), so
# these caches key on id(tensor) and evict via a weakref finalizer; entries
# double-check identity through the stored weakref to survive id reuse.
_CONFMAT_MUTATIONS: dict = {}
_CONFMAT_SCALARS_CACHE: dict = {}
_CONFMAT_SCRATCH: dict = {}


def _tensor_cache_get(cache: dict, t: Tensor):
    ent = cache.get(id(t))
    if ent is not None and ent[0]() is t:
        return ent[1]
    return None


def _tensor_cache_put(cache: dict, t: Tensor, value) -> None:
    key = id(t)

    def _evict(_ref, _key=key, _cache=cache):
        _cache.pop(_key, None)

    cache[key] = (_weakref.ref(t, _evict), value)


def _mark_kernel_mutated(t: Tensor) -> None:
    """Raw-pointer kernel writes bypass torch's _version counter; note them so
    version-keyed caches (confmat_scalars) invalidate correctly. Steady-state
    cost is one dict hit + in-place increment (this runs every update)."""
    ent = _CONFMAT_MUTATIONS.get(id(t))
    if ent is not None and ent[0]() is t:
        ent[1][0] += 1
        return
    _tensor_cache_put(_CONFMAT_MUTATIONS, t, [1])


def confmat_scalars(confmat: Tensor, zero_division: float = 0.0) -> Tensor:
    """Fused (C,C) confusion-matrix scalars: (mcc, unweighted kappa, macro
    jaccard) in two launches, cached per (tensor, version) so the metrics of a
    compute group (which alias one confmat state) pay for it once."""
    muts = _tensor_cache_get(_CONFMAT_MUTATIONS, confmat)
    ver = (confmat._version, muts[0] if muts is not None else 0)
    hit = _tensor_cache_get(_CONFMAT_SCALARS_CACHE, confmat)
    if hit is not None and hit[0] == ver and hit[2] == zero_division:
        return hit[1]
    lib = _lib()
    C = confmat.shape[0]
    dev = confmat.device
    key = (C, dev.index)
    scratch = _CONFMAT_SCRATCH.get(key)
    if scratch is None:
        scratch = torch.zeros(3 * C, dtype=torch.long, device=dev)
        if len(_CONFMAT_SCRATCH) > 16:
            _CONFMAT_SCRATCH.clear()
        _CONFMAT_SCRATCH[key] = scratch
    out = torch.empty(3, dtype=torch.float32, device=dev)
    cm = confmat if confmat.is_contiguous() else confmat.contiguous()
    rc = lib.ma_confmat_scalars(_stream(), cm.data_ptr(), C, scratch.data_ptr(),
                                float(zero_division), out.data_ptr())
    _check(rc, "ma_confmat_scalars")
    _tensor_cache_put(_CONFMAT_SCALARS_CACHE, confmat, (ver, out, zero_division))
    return out


def flush_curve_hist(owner) -> None:
    """Materialize an owner's lazily-accumulated curve histogram into its
    ``confmat`` state (suffix-sum; re-zeroes the histogram in-flight)."""
    meta = owner.__dict__.get("_hip_lazy_meta")
    buf = owner.__dict__.get("_hip_hist_buf")
    if meta is None or buf is None:
        return
    outer, T, transposed = meta
    confmat = owner.confmat
    assert confmat.is_contiguous() and confmat.device == buf.device
    rc = _lib().ma_curve_suffix(
        _stream(), buf.data_ptr(), outer, T, transposed, 1, 0, confmat.data_ptr()
    )
    _check(rc, "ma_curve_suffix")


def mc_stat_into(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int],
    scratch: Tensor, tp: Tensor, fp: Tensor, tn: Tensor, fn: Tensor,
) -> None:
    """Fused stat-scores update accumulated DIRECTLY into the metric states.

    ``scratch`` is a per-metric reusable (3*C+1,) int64 buffer; 2 launches
    total (count + single-block apply): the apply kernel consumes AND zeroes
    the whole scratch in one pass, so no fill kernel runs in steady state and
    no host-side epoch is needed (hipGraph-capturable).
    """
    lib = _lib()
    C = num_classes
    s_tp = scratch[:C]
    s_fp = scratch[C : 2 * C]
    s_fn = scratch[2 * C : 3 * C]
    s_valid_ptr = scratch.data_ptr() + 3 * C * 8
    if preds.ndim == 2 and preds.is_floating_point():
        preds = _to_supported(preds).contiguous()
        target = target.contiguous().long()
        B, C2 = preds.shape
        rc = lib.ma_mc_stat_logits(
            _stream(),
            preds.data_ptr(),
            _dtype_code(preds),
            target.data_ptr(),
            B,
            C2,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            s_tp.data_ptr(),
            s_fp.data_ptr(),
            s_fn.data_ptr(),
            0,
            s_valid_ptr,
            0,
            0, 0, 0,
        )
        _check(rc, "ma_mc_stat_logits")
    else:
        p2 = preds.contiguous().long().flatten()
        t2 = target.contiguous().long().flatten()
        rc = lib.ma_mc_stat_labels(
            _stream(),
            p2.data_ptr(),
            t2.data_ptr(),
            p2.numel(),
            C,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            s_tp.data_ptr(),
            s_fp.data_ptr(),
            s_fn.data_ptr(),
            0,
            s_valid_ptr,
        )
        _check(rc, "ma_mc_stat_labels")
    rc = lib.ma_apply_stat_deltas(
        _stream(),
        scratch.data_ptr(),
        C,
        tp.data_ptr(),
        fp.data_ptr(),
        tn.data_ptr(),
        fn.data_ptr(),
        0,
    )
    _check(rc, "ma_apply_stat_deltas")
    _mark_kernel_mutated(tp)


def mc_confmat_into(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int],
    confmat_state: Tensor, dummy: Tensor,
) -> None:
    """Confusion-matrix update atomically accumulated DIRECTLY into the (C,C)
    state — no per-batch confmat materialization, fill, or host-side add.

    ``dummy`` is a per-metric (3*C+1,) int64 sink for the per-class counters
    the kernel also produces (never read, never zeroed: garbage accumulates
    harmlessly in int64).
    """
    lib = _lib()
    C = num_classes
    assert confmat_state.is_contiguous()
    if preds.ndim == 2 and preds.is_floating_point():
        preds = _to_supported(preds).contiguous()
        target = target.contiguous().long()
        B, C2 = preds.shape
        rc = lib.ma_mc_stat_logits(
            _stream(),
            preds.data_ptr(),
            _dtype_code(preds),
            target.data_ptr(),
            B,
            C2,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            dummy[:C].data_ptr(),
            dummy[C : 2 * C].data_ptr(),
            dummy[2 * C : 3 * C].data_ptr(),
            confmat_state.data_ptr(),
            dummy[3 * C :].data_ptr(),
            0,
            0, 0, 0,
        )
        _check(rc, "ma_mc_stat_logits")
    else:
        p2 = preds.contiguous().long().flatten()
        t2 = target.contiguous().long().flatten()
        rc = lib.ma_mc_stat_labels(
            _stream(),
            p2.data_ptr(),
            t2.data_ptr(),
            p2.numel(),
            C,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            dummy[:C].data_ptr(),
            dummy[C : 2 * C].data_ptr(),
            dummy[2 * C : 3 * C].data_ptr(),
            confmat_state.data_ptr(),
            dummy[3 * C :].data_ptr(),
        )
        _check(rc, "ma_mc_stat_labels")
    _mark_kernel_mutated(confmat_state)


def mc_exact_into(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int],
    scratch: Tensor, correct: Tensor, total: Tensor,
) -> None:
    """Exact-match (== row argmax equals target) accumulated into correct/total.

    Reuses the fused stat kernel's per-class tp counts: correct = sum(tp),
    total = valid rows. Same (3*C+1,) self-zeroing scratch protocol as
    :func:`mc_stat_into`; 2 launches, zero fills in steady state.
    """
    lib = _lib()
    C = num_classes
    s_valid_ptr = scratch.data_ptr() + 3 * C * 8
    if preds.ndim == 2 and preds.is_floating_point():
        preds = _to_supported(preds).contiguous()
        target = target.contiguous().long()
        B, C2 = preds.shape
        rc = lib.ma_mc_stat_logits(
            _stream(),
            preds.data_ptr(),
            _dtype_code(preds),
            target.data_ptr(),
            B,
            C2,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            scratch[:C].data_ptr(),
            scratch[C : 2 * C].data_ptr(),
            scratch[2 * C : 3 * C].data_ptr(),
            0,
            s_valid_ptr,
            0,
            0, 0, 0,
        )
        _check(rc, "ma_mc_stat_logits")
    else:
        p2 = preds.contiguous().long().flatten()
        t2 = target.contiguous().long().flatten()
        rc = lib.ma_mc_stat_labels(
            _stream(),
            p2.data_ptr(),
            t2.data_ptr(),
            p2.numel(),
            C,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            scratch[:C].data_ptr(),
            scratch[C : 2 * C].data_ptr(),
            scratch[2 * C : 3 * C].data_ptr(),
            0,
            s_valid_ptr,
        )
        _check(rc, "ma_mc_stat_labels")
    rc = lib.ma_exact_apply(
        _stream(),
        scratch.data_ptr(),
        C,
        target.numel(),
        1,
        correct.data_ptr(),
        total.data_ptr(),
    )
    _check(rc, "ma_exact_apply")


# Batched linear-stat plans: the stat metrics of a compute group share one
# (tp,fp,tn,fn) state; each generation the FIRST caller batch-evaluates every
# formula seen so far in ONE kernel and later callers hit the cache. Keyed by
# id(tp) with weakref eviction; generation = (tp._version, kernel mutations).
_LINEAR_PLANS: dict = {}


def linear_stat_compute(
    tp: Tensor, fp: Tensor, tn: Tensor, fn: Tensor,
    num_coefs, den_coefs, average: str, zero_w_topk: bool = False,
    zero_division: float = 0.0, post_a: float = 1.0, post_b: float = 0.0,
) -> Tensor:
    """One-launch linear-ratio stat reduction (precision/recall/accuracy/...).

    score_c = post(safe_div(num_coefs . stats_c, den_coefs . stats_c)), then
    micro / macro / weighted averaging exactly like
    ``_adjust_weights_safe_divide``. Returns a 0-dim float32 tensor.
    Formulas over the same state tensor are batched per compute generation:
    one grid launch evaluates all of them (see ``k_linear_stat_multi``).
    """
    lib = _lib()
    C = tp.numel()
    avg_mode = {"macro": 0, "weighted": 1, "micro": 2}[average]
    key = (
        float(num_coefs[0]), float(num_coefs[1]), float(num_coefs[2]), float(num_coefs[3]),
        float(den_coefs[0]), float(den_coefs[1]), float(den_coefs[2]), float(den_coefs[3]),
        float(avg_mode), float(1 if zero_w_topk else 0),
        float(zero_division), float(post_a), float(post_b),
    )
    muts = _tensor_cache_get(_CONFMAT_MUTATIONS, tp)
    gen = (tp._version, muts[0] if muts is not None else 0)
    plan = _tensor_cache_get(_LINEAR_PLANS, tp)
    if plan is None:
        # formulas: key -> slot; params_dev rebuilt when the set grows
        plan = {"formulas": {}, "params_dev": None, "gen": None, "outs": None}
        _tensor_cache_put(_LINEAR_PLANS, tp, plan)
    slot = plan["formulas"].get(key)
    if slot is None:
        slot = len(plan["formulas"])
        plan["formulas"][key] = slot
        plan["params_dev"] = None  # set changed: re-upload next batch eval
        plan["gen"] = None
    if plan["gen"] == gen and plan["outs"] is not None:
        return plan["outs"][slot].reshape(())
    n = len(plan["formulas"])
    if plan["params_dev"] is None:
        flat = [0.0] * (13 * n)
        for k2, s2 in plan["formulas"].items():
            flat[13 * s2 : 13 * s2 + 13] = list(k2)
        plan["params_dev"] = torch.tensor(flat, dtype=torch.float32, device=tp.device)
    # fresh outs per generation: returned views must never be recycled
    outs = torch.empty(n, dtype=torch.float32, device=tp.device)
    rc = lib.ma_linear_stat_multi(
        _stream(), tp.data_ptr(), fp.data_ptr(), tn.data_ptr(), fn.data_ptr(),
        C, n, plan["params_dev"].data_ptr(), outs.data_ptr(),
    )
    _check(rc, "ma_linear_stat_multi")
    plan["gen"] = gen
    plan["outs"] = outs
    return outs[slot].reshape(())


def curve_auc_from_confmat(confmat_state: Tensor, mode: int) -> Tuple[Tensor, Tensor]:
    """Per-class AUROC (mode 0) / AveragePrecision (mode 1) + support weights
    straight from the (T, C, 2, 2) thresholded confmat state — one launch."""
    lib = _lib()
    assert confmat_state.ndim == 4 and confmat_state.is_contiguous()
    T, C = confmat_state.shape[0], confmat_state.shape[1]
    out = torch.empty(C, dtype=torch.float32, device=confmat_state.device)
    weights = torch.empty(C, dtype=torch.float32, device=confmat_state.device)
    rc = lib.ma_curve_auc_from_confmat(
        _stream(), confmat_state.data_ptr(), T, C, mode, out.data_ptr(), weights.data_ptr()
    )
    _check(rc, "ma_curve_auc_from_confmat")
    return out, weights


def mc_fused_collection_update(
    preds: Tensor, target: Tensor, num_classes: int, ignore_index: Optional[int],
    stat=None, confmat=None, exact=None, rowstats=None, defer_apply: bool = False,
    bump_epoch_ptr: int = 0,
):
    """Collection-level fused update: ONE pass over the (B, C) logits feeds the
    stat-scores, confusion-matrix and exact-match leaders at once.

    ``stat``  = (scratch, tp, fp, tn, fn) — required (owns the count scratch)
    ``confmat`` = confmat state tensor (atomics accumulate directly) or None
    ``exact`` = (correct, total) or None (reads the shared scratch BEFORE the
    apply kernel consumes and zeroes it).
    ``rowstats`` = (rowmax, rowinv, epoch_buf) or None: the same pass also
    emits the softmax row statistics + out-of-range epoch flag that the
    threshold-curve leader consumes (its own row-stats kernel is skipped).
    """
    lib = _lib()
    C = num_classes
    scratch, tp, fp, tn, fn = stat
    s_valid_ptr = scratch.data_ptr() + 3 * C * 8
    if preds.ndim == 2 and preds.is_floating_point():
        preds = _to_supported(preds).contiguous()
        target = target.contiguous().long()
        B = preds.shape[0]
        rc = lib.ma_mc_stat_logits(
            _stream(), preds.data_ptr(), _dtype_code(preds), target.data_ptr(),
            B, C,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            scratch[:C].data_ptr(), scratch[C : 2 * C].data_ptr(), scratch[2 * C : 3 * C].data_ptr(),
            confmat.data_ptr() if confmat is not None else 0,
            s_valid_ptr, 0,
            rowstats[0].data_ptr() if rowstats is not None else 0,
            rowstats[1].data_ptr() if rowstats is not None else 0,
            rowstats[2].data_ptr() if rowstats is not None else 0,
        )
        _check(rc, "ma_mc_stat_logits")
    else:
        p2 = preds.contiguous().long().flatten()
        t2 = target.contiguous().long().flatten()
        B = t2.numel()
        rc = lib.ma_mc_stat_labels(
            _stream(), p2.data_ptr(), t2.data_ptr(), B, C,
            ignore_index if ignore_index is not None else 0,
            1 if ignore_index is not None else 0,
            scratch[:C].data_ptr(), scratch[C : 2 * C].data_ptr(), scratch[2 * C : 3 * C].data_ptr(),
            confmat.data_ptr() if confmat is not None else 0,
            s_valid_ptr,
        )
        _check(rc, "ma_mc_stat_labels")
    def _apply_pass() -> None:
        # fused epilogue: stat-delta apply (+ exact-match) in ONE launch; when
        # the lazy curve hist rides the same stream the caller defers this so
        # it lands AFTER the hist kernel and closes the device epoch for free
        # (a separate 1-thread bump kernel costs a full ~4us dispatch)
        if exact is not None:
            correct, total = exact
            rc2 = lib.ma_apply_stat_exact(
                _stream(), scratch.data_ptr(), C, B,
                tp.data_ptr(), fp.data_ptr(), tn.data_ptr(), fn.data_ptr(),
                correct.data_ptr(), total.data_ptr(), bump_epoch_ptr,
            )
            _check(rc2, "ma_apply_stat_exact")
        else:
            rc2 = lib.ma_apply_stat_deltas(
                _stream(), scratch.data_ptr(), C,
                tp.data_ptr(), fp.data_ptr(), tn.data_ptr(), fn.data_ptr(), bump_epoch_ptr,
            )
            _check(rc2, "ma_apply_stat_deltas")

    if confmat is not None:
        _mark_kernel_mutated(confmat)
    _mark_kernel_mutated(tp)  # generation key for the batched linear-stat plan
    if defer_apply:
        return _apply_pass
    _apply_pass()
    return None
