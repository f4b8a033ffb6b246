"""rocTX tracing ranges around metric update/compute/sync.

SURVEY §5.1: the reference has no tracing subsystem; kernel-level visibility
is a goal here. When enabled (``METRICS_AMD_ROCTX=1`` or
``metrics_amd.utilities.tracing.enable()``), every metric's ``update``,
``compute`` and ``sync`` is wrapped in a named rocTX range so
``rocprofv3 --marker-trace`` attributes kernels to metrics.

Disabled (the default) this is a no-op with zero overhead on the hot path.
"""
from __future__ import annotations

import ctypes
import os
from contextlib import contextmanager, nullcontext
from typing import Iterator, Optional

_LIB: Optional[ctypes.CDLL] = None
_ENABLED = os.environ.get("METRICS_AMD_ROCTX", "0") == "1"


def _load() -> Optional[ctypes.CDLL]:
    global _LIB
    if _LIB is not None:
        return _LIB
    # the rocprofiler-SDK roctx is what rocprofv3 --marker-trace intercepts;
    # legacy libroctx64 ranges are invisible to it (verified on ROCm 7.x)
    for cand in (
        "/opt/rocm/lib/librocprofiler-sdk-roctx.so",
        "librocprofiler-sdk-roctx.so",
        "/opt/rocm/lib/libroctx64.so",
        "libroctx64.so",
    ):
        try:
            _LIB = ctypes.CDLL(cand)
            _LIB.roctxRangePushA.argtypes = [ctypes.c_char_p]
            return _LIB
        except OSError:
            continue
    return None


def enable() -> bool:
    """Turn on rocTX ranges; returns False if libroctx64 is unavailable."""
    global _ENABLED
    _ENABLED = _load() is not None
    return _ENABLED


def disable() -> None:
    global _ENABLED
    _ENABLED = False


def is_enabled() -> bool:
    return _ENABLED and _load() is not None


_NULL_CTX = nullcontext()


def range(name: str):  # noqa: A001
    """rocTX range context; no-op when tracing is disabled.

    Returns a shared ``nullcontext`` when disabled — a @contextmanager
    generator costs ~1-2us per entry, which matters at per-kernel call rates.
    """
    if not _ENABLED:
        return _NULL_CTX
    return _range_impl(name)


@contextmanager
def _range_impl(name: str) -> Iterator[None]:
    if not _ENABLED:
        yield
        return
    lib = _load()
    if lib is None:
        yield
        return
    lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        lib.roctxRangePop()
