"""Device-parity sweep: the modular differential case tables (originally
CPU-vs-reference) re-run with CUDA inputs, compared against the same metric
accumulating the same data on CPU. Broadens GPU coverage across every domain
the differential covers — catches device-placement and GPU-dispatch bugs the
dedicated kernel tests (which target specific kernels) can miss."""
import pytest
import torch

from tests.unittests.test_ref_differential_modular import (
    _ACC_CASES,
    _ACC_NS,
    _ACC_NS3,
    _ACC_NS5,
)

pytestmark = pytest.mark.gpu


def _to_dev(args, dev):
    out = []
    for a in args:
        if isinstance(a, torch.Tensor):
            out.append(a.to(dev))
        elif isinstance(a, dict):
            out.append({k: v.to(dev) for k, v in a.items()})
        else:
            return None  # non-tensor inputs (text) — not a device case
    return tuple(out)


def _run_pair(our_cls, kwargs, gen, atol):
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        cpu = our_cls(**kwargs)
        gpu = our_cls(**kwargs).to("cuda")
        for b in range(2):
            args = gen(90 + b)
            dev_args = _to_dev(args, "cuda")
            if dev_args is None:
                pytest.skip("non-tensor inputs")
            cpu.update(*args)
            gpu.update(*dev_args)
        a, b_ = cpu.compute(), gpu.compute()
        flat_a = a if isinstance(a, torch.Tensor) else torch.cat([x.flatten().float() for x in a])
        flat_b = b_ if isinstance(b_, torch.Tensor) else torch.cat([x.flatten().float().cpu() for x in b_])
        if isinstance(flat_b, torch.Tensor):
            flat_b = flat_b.cpu()
        assert torch.allclose(flat_a.float(), flat_b.float(), atol=max(atol, 1e-4), rtol=1e-3, equal_nan=True), (a, b_)


@pytest.mark.parametrize(
    ("name", "kwargs", "gen", "atol"), _ACC_CASES, ids=[f"{c[0]}_{i}" for i, c in enumerate(_ACC_CASES)]
)
def test_device_parity(name, kwargs, gen, atol):
    import metrics_amd as ma

    _run_pair(getattr(ma, name), kwargs, gen, atol)


_NS_ALL = [c for c in (_ACC_NS + _ACC_NS3 + _ACC_NS5)]


@pytest.mark.parametrize(
    ("ns", "name", "kwargs", "gen", "atol"), _NS_ALL, ids=[f"{c[1]}_{i}" for i, c in enumerate(_NS_ALL)]
)
def test_device_parity_ns(ns, name, kwargs, gen, atol):
    import importlib

    our_ns = importlib.import_module(f"metrics_amd.{ns}")
    _run_pair(getattr(our_ns, name), kwargs, gen, atol)
