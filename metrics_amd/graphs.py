"""hipGraph-captured metric updates.

A 16-metric collection update is ~15 kernels; at ~0.3 ms/step roughly a third
of the wall time is host launch overhead and Python between those launches.
``GraphedUpdate`` captures one ``update()`` into a hipGraph (``torch.cuda.
CUDAGraph`` is hipGraph on ROCm) and replays it per step: one graph launch
replaces the whole launch train.

Requirements (all true for the fused MI355X update paths):
- every state is a fixed-shape tensor (no list/``cat`` states),
- the update makes no host<->device sync and allocates no new tensors after
  warmup (the kernel scratch/histogram pools are persistent),
- no host-side state feeds kernel arguments (the out-of-range flag protocol
  keeps its epoch ON DEVICE — see csrc/kernels.hip k_curve_suffix).

Typical use::

    coll = ma.MetricCollection({...}).to("cuda")
    graphed = ma.graphs.GraphedUpdate(coll, example_preds, example_target)
    for batch in loader:
        graphed.update(batch.preds, batch.target)   # one graph replay
    coll.compute()                                   # unchanged (incl. DDP sync)
"""
from __future__ import annotations

from typing import Any, Iterable

import torch
from torch import Tensor

from metrics_amd.metric import Metric

__all__ = ["GraphedUpdate"]


def _metrics_of(target: Any) -> Iterable[Metric]:
    if isinstance(target, Metric):
        return [target]
    # MetricCollection: iterate base metrics without compute-group copies
    return list(target.values(copy_state=False))


class GraphedUpdate:
    """Capture ``target.update(*example_args)`` into a hipGraph and replay it.

    ``target`` is a :class:`~metrics_amd.metric.Metric` or ``MetricCollection``
    whose states are all fixed-shape tensors. After construction the states
    are reset in place (warmup pollution removed) and each :meth:`update`
    copies the batch into the static input buffers and replays the graph.
    """

    def __init__(self, target: Any, *example_args: Tensor, warmup: int = 3, parallel: bool = True) -> None:
        if not torch.cuda.is_available():
            raise RuntimeError("GraphedUpdate requires a GPU (hipGraph capture)")
        self.target = target
        self._metrics = list(_metrics_of(target))
        for m in self._metrics:
            for name in m._defaults:
                if isinstance(getattr(m, name), list):
                    raise RuntimeError(
                        f"GraphedUpdate needs fixed-shape tensor states, but {m.__class__.__name__}.{name}"
                        " is a list state (its shape grows per update)."
                    )
        self._static = tuple(a.detach().clone() for a in example_args)

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                target.update(*self._static)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        # compute-group leaders are independent (disjoint states, read-only
        # inputs, per-metric kernel scratch): record each on its own stream so
        # replay overlaps their kernels — the graph becomes a parallel DAG
        leaders = None
        if parallel and not isinstance(target, Metric) and getattr(target, "_groups_checked", False):
            leaders = [getattr(target, members[0]) for members in target._groups.values()]
        self._streams = [torch.cuda.Stream() for _ in (leaders or [])]

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            if leaders and len(leaders) > 1:
                main = torch.cuda.current_stream()
                for m, st in zip(leaders, self._streams):
                    st.wait_stream(main)
                    with torch.cuda.stream(st):
                        m.update(*self._static)
                for st in self._streams:
                    main.wait_stream(st)
            else:
                target.update(*self._static)

        # capture records but does not execute: only the warmup polluted the
        # states — restore defaults IN PLACE (the graph holds state pointers)
        self._reset_states_inplace()
        for m in self._metrics:
            m._update_count = 0
            m._computed = None

    def _reset_states_inplace(self) -> None:
        for m in self._metrics:
            for name, default in m._defaults.items():
                state = getattr(m, name)
                if isinstance(state, Tensor):
                    state.copy_(default.to(state.device))
            # drop lazily-accumulated curve histograms with the states
            buf = m.__dict__.get("_hip_hist_buf")
            if buf is not None:
                buf.zero_()
            m.__dict__["_lazy_dirty"] = False

    def reset_states(self) -> None:
        """In-place equivalent of ``target.reset()`` (``reset()`` allocates new
        state tensors, which would orphan the graph's captured pointers)."""
        self._reset_states_inplace()
        for m in self._metrics:
            m._update_count = 0
            m._computed = None

    def update(self, *args: Tensor) -> None:
        """Copy the batch into the static buffers and replay the captured graph."""
        for buf, a in zip(self._static, args):
            buf.copy_(a, non_blocking=True)
        self.graph.replay()
        for m in self._metrics:
            m._update_count += 1
            m._computed = None
            # replays bypass Python update(): re-arm the lazy-curve dirty flag
            # so the next state read materializes the accumulated histogram
            if "_hip_lazy_meta" in m.__dict__:
                m.__dict__["_lazy_dirty"] = True
        # copy-on-access (items()/values()) de-aliases compute-group members;
        # re-establish the state refs so compute() sees the leaders' states
        t = self.target
        if getattr(t, "_state_is_copy", False):
            t._compute_groups_create_state_ref()
            t._state_is_copy = False

    __call__ = update
