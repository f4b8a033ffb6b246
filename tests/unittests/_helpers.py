"""Test helpers: oracle-comparison runner + 2-process gloo cluster emulation.

Modeled on the reference's MetricTester strategy (torchmetrics
tests/unittests/_helpers/testers.py): batch-by-batch forward vs reference,
final compute vs reference on all data, clone/pickle/reset checks; DDP tests
run the same check over CPU gloo processes with interleaved batches.
"""
from __future__ import annotations

import os
import pickle
from functools import partial
from typing import Any, Callable, Dict, Optional, Sequence

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import Tensor

from metrics_amd.metric import Metric

NUM_BATCHES = 4
BATCH_SIZE = 32
NUM_CLASSES = 5


def seed_all(seed: int = 42) -> None:
    np.random.seed(seed)
    torch.manual_seed(seed)


def _assert_allclose(res: Any, ref: Any, atol: float = 1e-5) -> None:
    if isinstance(res, Tensor):
        ref_t = torch.as_tensor(np.asarray(ref), dtype=torch.float64)
        assert torch.allclose(res.double().cpu(), ref_t, atol=atol), f"{res} vs {ref}"
    elif isinstance(res, dict):
        for k in res:
            _assert_allclose(res[k], ref[k], atol)
    elif isinstance(res, (list, tuple)):
        for r, rf in zip(res, ref):
            _assert_allclose(r, rf, atol)
    else:
        assert abs(float(res) - float(ref)) < atol


def run_class_metric_test(
    metric_class: Callable,
    ref_fn: Callable,
    preds: Tensor,
    target: Tensor,
    metric_args: Optional[Dict[str, Any]] = None,
    check_batch: bool = True,
    atol: float = 1e-5,
) -> None:
    """Instantiate, run batch-by-batch forward + final compute vs the oracle.

    ``preds``/``target`` are (NUM_BATCHES, B, ...); ``ref_fn(preds_np, target_np)``
    computes the expected value over concatenated data.
    """
    metric_args = metric_args or {}
    metric = metric_class(**metric_args)

    # pickle roundtrip must work
    metric = pickle.loads(pickle.dumps(metric))
    cloned = metric.clone()
    assert cloned is not metric

    for i in range(preds.shape[0]):
        batch_val = metric(preds[i], target[i])
        if check_batch:
            ref_b = ref_fn(preds[i], target[i])
            _assert_allclose(batch_val, ref_b, atol)

    total_val = metric.compute()
    flat_p = preds.reshape(-1, *preds.shape[2:])
    flat_t = target.reshape(-1, *target.shape[2:])
    ref_total = ref_fn(flat_p, flat_t)
    _assert_allclose(total_val, ref_total, atol)

    # reset restores defaults
    metric.reset()
    for name, default in metric._defaults.items():
        cur = getattr(metric, name)
        if isinstance(default, Tensor):
            assert torch.allclose(cur, default.to(cur.device))
        else:
            assert cur == []


def run_functional_metric_test(
    metric_fn: Callable,
    ref_fn: Callable,
    preds: Tensor,
    target: Tensor,
    metric_args: Optional[Dict[str, Any]] = None,
    atol: float = 1e-5,
) -> None:
    metric_args = metric_args or {}
    for i in range(preds.shape[0]):
        res = metric_fn(preds[i], target[i], **metric_args)
        ref = ref_fn(preds[i], target[i])
        _assert_allclose(res, ref, atol)


def run_class_metric_ddp_test(
    metric_class: Callable,
    ref_fn: Callable,
    preds: Tensor,
    target: Tensor,
    metric_args: Optional[Dict[str, Any]] = None,
    atol: float = 1e-5,
    world_size: int = 2,
) -> None:
    """DDP pool sweep (reference _class_test ddp mode): each rank consumes
    interleaved batches; the synced compute must equal the oracle on ALL data."""

    def worker(rank: int, ws: int) -> None:
        m = metric_class(**(metric_args or {}))
        for i in range(rank, preds.shape[0], ws):
            m.update(preds[i], target[i])
        val = m.compute()
        flat_p = preds.reshape(-1, *preds.shape[2:])
        flat_t = target.reshape(-1, *target.shape[2:])
        _assert_allclose(val, ref_fn(flat_p, flat_t), atol)

    run_distributed(worker, world_size=world_size)


def run_dtype_test(
    metric_class: Callable,
    preds: Tensor,
    target: Tensor,
    metric_args: Optional[Dict[str, Any]] = None,
    dtype: torch.dtype = torch.double,
    atol: float = 1e-4,
) -> None:
    """Reference run_precision_test: the metric computed with states/inputs in
    ``dtype`` must agree with the fp32 run."""
    base = metric_class(**(metric_args or {}))
    for i in range(preds.shape[0]):
        base.update(preds[i], target[i])
    expected = base.compute()

    m = metric_class(**(metric_args or {})).set_dtype(dtype)
    p = preds.to(dtype) if preds.is_floating_point() else preds
    for i in range(p.shape[0]):
        m.update(p[i], target[i])
    _assert_allclose(m.compute(), expected, atol)


# ---------------------------------------------------------------- distributed
def _dist_worker(rank: int, world_size: int, port: int, fn: Callable, args: tuple, backend: str = "gloo") -> None:
    # Forked children inherit the parent's (possibly mid-operation) OpenMP pool
    # state; entering a parallel region then deadlocks on a stale futex. Keep
    # the child single-threaded so no parallel region is ever entered.
    torch.set_num_threads(1)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    on_gpu = backend in ("nccl", "gloo_cuda")
    if on_gpu:
        # single-box proof: every rank shares cuda:0
        torch.cuda.set_device(0)
    dist.init_process_group("gloo" if backend == "gloo_cuda" else backend, rank=rank, world_size=world_size)
    try:
        fn(rank, world_size, *args)
        if on_gpu:
            torch.cuda.synchronize()
            dist.barrier()
    finally:
        dist.destroy_process_group()


def _free_port() -> int:
    """Ask the OS for a free TCP port (avoids RNG-driven collisions across tests)."""
    import socket

    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_distributed(fn: Callable, world_size: int = 2, args: tuple = (), backend: str = "gloo") -> None:
    """Run ``fn(rank, world_size, *args)`` in ``world_size`` processes on localhost.

    backend "gloo" = CPU cluster emulation (fork). backend "nccl" = real RCCL.
    backend "gloo_cuda" = gloo group with ranks pinned to cuda:0 (multi-rank
    reductions of CUDA states on a 1-GPU box; gloo stages via host). GPU
    backends use spawn — fork is unsafe after HIP init.
    """
    port = _free_port()
    on_gpu = backend in ("nccl", "gloo_cuda")
    mp.start_processes(
        partial(_dist_worker, world_size=world_size, port=port, fn=fn, args=args, backend=backend),
        nprocs=world_size,
        start_method="spawn" if (on_gpu or os.environ.get("MA_DIST_SPAWN")) else "fork",
    )
