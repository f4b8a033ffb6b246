"""L2 — the core stateful Metric runtime.

Parity target: torchmetrics ``metric.py`` (Metric ABC + CompositionalMetric):
add_state / update / compute / forward / reset / sync / unsync / sync_context /
merge_state / state_dict / clone / plot, with byte-compatible state-dict layout
(``prefix + state_name`` keys, persistent-gated).

MI355X-native deltas:
- distributed sync maps reductions onto RCCL collectives (fused all-reduce for
  sum/mean/max/min states, all-gather only for cat/None/custom) — see
  ``utilities/distributed.py``; the reference gathers everything.
- no TorchScript support, no XLA/MPS/deterministic fallback ladders: the
  compute path is PyTorch-ROCm + in-tree HIP kernels (gfx950) only.
"""
from __future__ import annotations

import functools
import inspect
from abc import ABC, abstractmethod
from contextlib import contextmanager
from copy import deepcopy
from typing import Any, Callable, Dict, Generator, List, Optional, Sequence, Union

import torch
from torch import Tensor
from torch.nn import Module

from metrics_amd.utilities.data import (
    _flatten,
    apply_to_collection,
    dim_zero_cat,
    dim_zero_max,
    dim_zero_mean,
    dim_zero_min,
    dim_zero_sum,
)
from metrics_amd.utilities.distributed import gather_all_tensors, sync_states_fast
from metrics_amd.utilities.exceptions import MetricsUserError
from metrics_amd.utilities.prints import rank_zero_warn
from metrics_amd.utilities import tracing


def jit_distributed_available() -> bool:
    import torch.distributed as dist

    return dist.is_available() and dist.is_initialized()


_BUILTIN_REDUCTIONS = {
    "sum": dim_zero_sum,
    "mean": dim_zero_mean,
    "cat": dim_zero_cat,
    "min": dim_zero_min,
    "max": dim_zero_max,
}


def _squeeze_if_scalar(data: Any) -> Any:
    return apply_to_collection(data, Tensor, lambda x: x.squeeze() if x.numel() == 1 else x)


def _clone_result(value: Any, state_storages: Optional[frozenset] = None) -> Any:
    """Detach compute() results from the metric states.

    The contract only requires the result not to alias metric state. Most of
    our fused compute paths return freshly allocated tensors, so when
    ``state_storages`` (the states' storage data_ptrs) is given, results whose
    storage is not a state storage are returned AS-IS — each skipped clone is
    a saved kernel dispatch (~4us each; compute is dispatch-bound).
    Per-class curve results are lists of many small GPU tensors; cloning them
    one by one costs one D2D launch each (~100 launches per compute on the
    curve metrics — profiles/README.md prof15). Batch same-(device,dtype)
    tensors through ONE cat and hand back views of the packed buffer.
    """
    if state_storages is not None:
        def _aliases_state(t: Tensor) -> bool:
            try:
                return t.untyped_storage().data_ptr() in state_storages
            except Exception:
                return True

        any_alias = False

        def _scan(t: Tensor) -> Tensor:
            nonlocal any_alias
            any_alias = any_alias or _aliases_state(t)
            return t

        apply_to_collection(value, Tensor, _scan)
        if not any_alias:
            return value
    tensors: List[Tensor] = []

    def _collect(t: Tensor) -> Tensor:
        tensors.append(t)
        return t

    apply_to_collection(value, Tensor, _collect)
    n_cuda = sum(1 for t in tensors if t.is_cuda)
    if n_cuda < 4:
        return apply_to_collection(value, Tensor, lambda x: x.clone())

    groups: Dict[Any, List[Tensor]] = {}
    for t in tensors:
        groups.setdefault((t.device, t.dtype), []).append(t)
    cloned: Dict[int, Tensor] = {}
    for (dev, _dt), ts in groups.items():
        if dev.type != "cuda" or len(ts) == 1:
            for t in ts:
                cloned.setdefault(id(t), t.clone())
        else:
            uniq = []
            seen = set()
            for t in ts:
                if id(t) not in seen:
                    seen.add(id(t))
                    uniq.append(t)
            flat = torch.cat([t.reshape(-1) for t in uniq])
            off = 0
            for t in uniq:
                n = t.numel()
                cloned[id(t)] = flat[off : off + n].view(t.shape)
                off += n
    return apply_to_collection(value, Tensor, lambda x: cloned[id(x)])


class Metric(Module, ABC):
    """Base class for all metrics.

    Subclasses declare accumulator states with :meth:`add_state` and implement
    ``update(...)`` and ``compute()``. The base class provides batch/global
    ``forward`` semantics, RCCL-backed distributed synchronization, resets,
    checkpointing and operator composition.
    """

    __jit_ignored_attributes__: List[str] = ["device", "dtype"]
    is_differentiable: Optional[bool] = None
    higher_is_better: Optional[bool] = None
    full_state_update: Optional[bool] = None
    plot_lower_bound: Optional[float] = None
    plot_upper_bound: Optional[float] = None
    plot_legend_name: Optional[str] = None

    def __init__(self, **kwargs: Any) -> None:
        super().__init__()

        self._device = torch.device("cpu")
        self._dtype = torch.get_default_dtype()

        self.compute_on_cpu = kwargs.pop("compute_on_cpu", False)
        if not isinstance(self.compute_on_cpu, bool):
            raise ValueError(f"Expected keyword argument `compute_on_cpu` to be a `bool` but got {self.compute_on_cpu}")

        self.dist_sync_on_step = kwargs.pop("dist_sync_on_step", False)
        if not isinstance(self.dist_sync_on_step, bool):
            raise ValueError(
                f"Expected keyword argument `dist_sync_on_step` to be a `bool` but got {self.dist_sync_on_step}"
            )

        self.process_group = kwargs.pop("process_group", None)

        self.dist_sync_fn = kwargs.pop("dist_sync_fn", None)
        if self.dist_sync_fn is not None and not callable(self.dist_sync_fn):
            raise ValueError(
                f"Expected keyword argument `dist_sync_fn` to be a callable or None but got {self.dist_sync_fn}"
            )

        self.distributed_available_fn = kwargs.pop("distributed_available_fn", None) or jit_distributed_available

        self.sync_on_compute = kwargs.pop("sync_on_compute", True)
        if not isinstance(self.sync_on_compute, bool):
            raise ValueError(
                f"Expected keyword argument `sync_on_compute` to be a `bool` but got {self.sync_on_compute}"
            )
        self.compute_with_cache = kwargs.pop("compute_with_cache", True)
        if not isinstance(self.compute_with_cache, bool):
            raise ValueError(
                f"Expected keyword argument `compute_with_cache` to be a `bool` but got {self.compute_with_cache}"
            )

        if kwargs:
            kwargs_ = [f"`{a}`" for a in sorted(kwargs)]
            raise ValueError(f"Unexpected keyword arguments: {', '.join(kwargs_)}")

        # state management
        self._update_signature = inspect.signature(self.update)
        self.update: Callable = self._wrap_update(self.update)  # type: ignore[method-assign]
        self.compute: Callable = self._wrap_compute(self.compute)  # type: ignore[method-assign]
        self._computed: Any = None
        self._forward_cache: Any = None
        self._update_count = 0
        self._to_sync = self.sync_on_compute
        self._should_unsync = True
        self._enable_grad = False
        self._dtype_convert = False

        self._defaults: Dict[str, Union[List, Tensor]] = {}
        self._persistent: Dict[str, bool] = {}
        self._reductions: Dict[str, Union[str, Callable, None]] = {}

        self._is_synced = False
        self._cache: Optional[Dict[str, Union[List[Tensor], Tensor]]] = None
        self._pending_sync_event: Optional[Any] = None

    # ------------------------------------------------------------------ props
    @property
    def _update_called(self) -> bool:
        return self._update_count > 0

    @property
    def update_called(self) -> bool:
        """True if ``update`` or ``forward`` has been called at least once."""
        return self._update_count > 0

    @property
    def update_count(self) -> int:
        """Number of times ``update``/``forward`` has been called."""
        return self._update_count

    @property
    def metric_state(self) -> Dict[str, Union[List[Tensor], Tensor]]:
        """Current values of all registered metric states."""
        return {attr: getattr(self, attr) for attr in self._defaults}

    @property
    def device(self) -> torch.device:
        """Device of the metric states."""
        return self._device

    @property
    def dtype(self) -> torch.dtype:
        """Dtype of the metric states."""
        return self._dtype

    # ------------------------------------------------------------------ state
    def add_state(
        self,
        name: str,
        default: Union[list, Tensor],
        dist_reduce_fx: Optional[Union[str, Callable]] = None,
        persistent: bool = False,
    ) -> None:
        """Register an accumulator state.

        ``dist_reduce_fx`` in {'sum','mean','cat','min','max'}, a callable, or
        None. sum/mean/min/max states sync with a fused RCCL all-reduce;
        cat/None/custom with a shape-exchanging all-gather.
        """
        if not isinstance(default, (Tensor, list)) or (isinstance(default, list) and default):
            raise ValueError("state variable must be a tensor or any empty list (where you can append tensors)")
        if dist_reduce_fx is not None and not (dist_reduce_fx in _BUILTIN_REDUCTIONS or callable(dist_reduce_fx)):
            raise ValueError("`dist_reduce_fx` must be callable or one of ['mean', 'sum', 'cat', 'min', 'max', None]")
        if isinstance(default, Tensor):
            default = default.contiguous()

        setattr(self, name, default)
        self._defaults[name] = deepcopy(default)
        self._persistent[name] = persistent
        self._reductions[name] = dist_reduce_fx

    # ---------------------------------------------------------------- forward
    def _state_storages(self) -> frozenset:
        ptrs = []
        for key in self._defaults:
            val = getattr(self, key, None)
            if isinstance(val, Tensor):
                try:
                    ptrs.append(val.untyped_storage().data_ptr())
                except Exception:
                    pass
            elif isinstance(val, list):
                for v in val:
                    if isinstance(v, Tensor):
                        try:
                            ptrs.append(v.untyped_storage().data_ptr())
                        except Exception:
                            pass
        return frozenset(ptrs)

    def _lazy_flush(self) -> None:
        """Materialize deferred device-side state (lazy curve histograms).

        Metrics on the lazy GPU curve path accumulate raw histograms per
        update and defer the suffix-sum/confmat materialization to the first
        state read (compute / sync / state_dict / forward / device move).
        """
        from metrics_amd.ops import _hip

        _hip.flush_curve_hist(self)

    def _maybe_flush_lazy(self) -> None:
        if self.__dict__.get("_lazy_dirty"):
            self.__dict__["_lazy_dirty"] = False
            self._lazy_flush()

    def forward(self, *args: Any, **kwargs: Any) -> Any:
        """Accumulate the batch into the global state AND return the batch value."""
        self._maybe_flush_lazy()
        if self.full_state_update or self.full_state_update is None or self.dist_sync_on_step:
            self._forward_cache = self._forward_full_state_update(*args, **kwargs)
        else:
            self._forward_cache = self._forward_reduce_state_update(*args, **kwargs)
        return self._forward_cache

    def _forward_full_state_update(self, *args: Any, **kwargs: Any) -> Any:
        """Safe path: two update calls (global accumulate + fresh batch state)."""
        self.update(*args, **kwargs)
        _update_count = self._update_count

        self._to_sync = self.dist_sync_on_step
        self._should_unsync = False
        _temp_compute_on_cpu = self.compute_on_cpu
        self.compute_on_cpu = False

        cache = self._copy_state_dict()

        self.reset()
        self._enable_grad = True
        self.update(*args, **kwargs)
        batch_val = self.compute()

        # restore global state
        for attr, val in cache.items():
            setattr(self, attr, val)
        self._update_count = _update_count
        self._restore_forward_context(_temp_compute_on_cpu)
        return batch_val

    def _forward_reduce_state_update(self, *args: Any, **kwargs: Any) -> Any:
        """Fast path: one update on a fresh state, then a pairwise state merge."""
        global_state = self._copy_state_dict()
        _update_count = self._update_count
        self.reset()

        self._to_sync = self.dist_sync_on_step
        self._should_unsync = False
        _temp_compute_on_cpu = self.compute_on_cpu
        self.compute_on_cpu = False
        self._enable_grad = True

        self.update(*args, **kwargs)
        batch_val = self.compute()

        self._update_count = _update_count + 1
        with torch.no_grad():
            self._reduce_states(global_state)

        self._restore_forward_context(_temp_compute_on_cpu)
        return batch_val

    def _restore_forward_context(self, _temp_compute_on_cpu: bool) -> None:
        self._is_synced = False
        self._should_unsync = True
        self._to_sync = self.sync_on_compute
        self._computed = None
        self._enable_grad = False
        self.compute_on_cpu = _temp_compute_on_cpu
        if self.compute_on_cpu:
            self._move_list_states_to_cpu()

    # ----------------------------------------------------------- state merges
    def _reduce_states(self, incoming_state: Dict[str, Any]) -> None:
        """Merge ``incoming_state`` (a prior/global state) into the current state."""
        for attr in self._defaults:
            local_state = getattr(self, attr)
            if attr not in incoming_state:
                continue
            global_state = incoming_state[attr]
            reduce_fn = self._reductions[attr]
            if reduce_fn == "sum":
                reduced = global_state + local_state
            elif reduce_fn == "mean":
                reduced = ((self._update_count - 1) * global_state + local_state).float() / self._update_count
            elif reduce_fn == "max":
                reduced = torch.max(global_state, local_state)
            elif reduce_fn == "min":
                reduced = torch.min(global_state, local_state)
            elif reduce_fn == "cat":
                reduced = global_state + local_state if isinstance(local_state, list) else torch.cat(
                    [global_state, local_state]
                )
            elif reduce_fn is None and isinstance(global_state, Tensor):
                reduced = torch.stack([global_state, local_state])
            elif reduce_fn is None and isinstance(global_state, list):
                reduced = _flatten([global_state, local_state])
            elif callable(reduce_fn):
                reduced = reduce_fn(torch.stack([global_state, local_state]))
            else:
                raise TypeError(f"Unsupported reduce_fn: {reduce_fn}")
            setattr(self, attr, reduced)

    def merge_state(self, incoming_state: Union[Dict[str, Any], "Metric"]) -> None:
        """Merge an external metric state (another instance or a state dict) into this one."""
        self._maybe_flush_lazy()
        if isinstance(incoming_state, Metric):
            incoming_state._maybe_flush_lazy()
        if not isinstance(incoming_state, (dict, Metric)):
            raise ValueError(
                f"Expected incoming state to be a dict or an instance of Metric but got {type(incoming_state)}"
            )
        if self.full_state_update or self.full_state_update is None or self.dist_sync_on_step:
            raise RuntimeError(
                "``merge_state`` is not supported for metrics with ``full_state_update=True`` "
                "or ``dist_sync_on_step=True``"
            )
        if isinstance(incoming_state, Metric):
            if self.__class__.__name__ != incoming_state.__class__.__name__:
                raise ValueError(
                    f"Expected incoming state to be an instance of {self.__class__.__name__} "
                    f"but got {incoming_state.__class__.__name__}"
                )
            count_increment = incoming_state._update_count
            incoming_state = {attr: getattr(incoming_state, attr) for attr in incoming_state._defaults}
        else:
            unknown_keys = [k for k in incoming_state if k not in self._defaults]
            if unknown_keys:
                raise RuntimeError(f"Found unknown key(s) in incoming state: {unknown_keys}")
            count_increment = 1

        self._update_count += count_increment
        self._computed = None  # merged state invalidates any cached result
        with torch.no_grad():
            self._reduce_states(incoming_state)

    # ------------------------------------------------------------------- sync
    def _sync_dist(self, dist_sync_fn: Callable = gather_all_tensors, process_group: Optional[Any] = None) -> None:
        states = {attr: getattr(self, attr) for attr in self._reductions}
        kinds: Dict[str, str] = {}
        customs: Dict[str, Callable] = {}
        for name, fn in self._reductions.items():
            if isinstance(fn, str):
                kinds[name] = fn
            elif fn is None:
                kinds[name] = "none"
            else:
                kinds[name] = "custom"
                customs[name] = fn
        gather_fn = None if dist_sync_fn in (None, gather_all_tensors) else dist_sync_fn
        synced, done_event = sync_states_fast(
            states, kinds, customs, group=process_group or self.process_group, gather_fn=gather_fn
        )
        for attr, val in synced.items():
            setattr(self, attr, val)
        # collectives ran on the side HIP stream; whoever reads the synced
        # states (compute) waits on this lazily — see _wait_pending_sync
        self._pending_sync_event = done_event

    def _wait_pending_sync(self) -> None:
        """Fence the current stream behind an in-flight side-stream sync."""
        ev = getattr(self, "_pending_sync_event", None)
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)
            # pin the side-stream-allocated result buffers to this stream so
            # the caching allocator cannot hand them out early
            for attr in self._defaults:
                val = getattr(self, attr)
                if isinstance(val, Tensor) and val.is_cuda:
                    val.record_stream(torch.cuda.current_stream())
            self._pending_sync_event = None

    def sync(
        self,
        dist_sync_fn: Optional[Callable] = None,
        process_group: Optional[Any] = None,
        should_sync: bool = True,
        distributed_available: Optional[Callable] = None,
    ) -> None:
        """Synchronize metric states across processes (caches local state for :meth:`unsync`)."""
        self._maybe_flush_lazy()
        if self._is_synced and should_sync:
            raise MetricsUserError("The Metric has already been synced.")

        if distributed_available is None and self.distributed_available_fn is not None:
            distributed_available = self.distributed_available_fn
        is_distributed = distributed_available() if callable(distributed_available) else None

        if not should_sync or not is_distributed:
            return

        if dist_sync_fn is None:
            dist_sync_fn = gather_all_tensors

        # cache prior to syncing
        self._cache = self._copy_state_dict()

        # sync
        with tracing.range(f"{self.__class__.__name__}.sync"):
            self._sync_dist(dist_sync_fn, process_group=process_group)
        self._is_synced = True

    def unsync(self, should_unsync: bool = True) -> None:
        """Restore the cached local (pre-sync) state."""
        if not should_unsync:
            return
        if not self._is_synced:
            raise MetricsUserError("The Metric has already been un-synced.")
        if self._cache is None:
            raise MetricsUserError("The internal cache should exist to unsync the Metric.")

        for attr, val in self._cache.items():
            setattr(self, attr, val)
        self._is_synced = False
        self._cache = None
        # a still-in-flight side-stream sync only touches its own buffers,
        # which we just dropped — nothing left to wait for
        self._pending_sync_event = None

    @contextmanager
    def sync_context(
        self,
        dist_sync_fn: Optional[Callable] = None,
        process_group: Optional[Any] = None,
        should_sync: bool = True,
        should_unsync: bool = True,
        distributed_available: Optional[Callable] = None,
    ) -> Generator:
        """Context manager: synced states inside, local states restored on exit."""
        self.sync(
            dist_sync_fn=dist_sync_fn,
            process_group=process_group,
            should_sync=should_sync,
            distributed_available=distributed_available,
        )
        yield
        self.unsync(should_unsync=self._is_synced and should_unsync)

    # --------------------------------------------------------------- wrapping
    def _wrap_update(self, update: Callable) -> Callable:
        @functools.wraps(update)
        def wrapped_func(*args: Any, **kwargs: Any) -> None:
            self._computed = None
            self._update_count += 1
            with torch.set_grad_enabled(self._enable_grad), tracing.range(f"{self.__class__.__name__}.update"):
                try:
                    update(*args, **kwargs)
                except RuntimeError as err:
                    if "Expected all tensors to be on" in str(err):
                        raise RuntimeError(
                            "Encountered different devices in metric calculation (see stacktrace for details). "
                            "This could be due to the metric class not being on the same device as input. "
                            f"Instead of `metric={self.__class__.__name__}(...)` try to do "
                            f"`metric={self.__class__.__name__}(...).to(device)` where "
                            "device corresponds to the device of the input."
                        ) from err
                    raise err

            if self.compute_on_cpu:
                self._move_list_states_to_cpu()

        return wrapped_func

    def _move_list_states_to_cpu(self) -> None:
        """Move list states to cpu to save GPU memory."""
        for key in self._defaults:
            current_val = getattr(self, key)
            if isinstance(current_val, Sequence):
                setattr(self, key, [cur_v.to("cpu") for cur_v in current_val])

    def _wrap_compute(self, compute: Callable) -> Callable:
        @functools.wraps(compute)
        def wrapped_func(*args: Any, **kwargs: Any) -> Any:
            self._maybe_flush_lazy()
            if self._update_count == 0:
                rank_zero_warn(
                    f"The ``compute`` method of metric {self.__class__.__name__}"
                    " was called before the ``update`` method which may lead to errors,"
                    " as metric states have not yet been updated.",
                    UserWarning,
                )

            if self._computed is not None:
                return self._computed

            # fast path: nothing to sync (single process / sync disabled),
            # no pending side-stream event, tracing off — skip the sync/unsync
            # context machinery entirely (two generator contexts per metric
            # per compute are pure dispatch overhead in the hot loop)
            if (
                not self._is_synced
                and self.__dict__.get("_pending_sync_event") is None
                and not (self._to_sync and self.distributed_available_fn())
                and not tracing.is_enabled()
            ):
                value = _squeeze_if_scalar(compute(*args, **kwargs))
                value = _clone_result(value, self._state_storages())
                if self.compute_with_cache:
                    self._computed = value
                return value

            with self.sync_context(
                dist_sync_fn=self.dist_sync_fn,
                should_sync=self._to_sync,
                should_unsync=self._should_unsync,
            ), tracing.range(f"{self.__class__.__name__}.compute"):
                self._wait_pending_sync()
                value = _squeeze_if_scalar(compute(*args, **kwargs))
                # detach from state so later in-place ops cannot alter the
                # returned result (storage-checked: fresh allocations pass
                # through; aliases of state get the batched clone)
                value = _clone_result(value, self._state_storages())

            if self.compute_with_cache:
                self._computed = value
            return value

        return wrapped_func

    # --------------------------------------------------------------- abstract
    @abstractmethod
    def update(self, *_: Any, **__: Any) -> None:
        """Override: accumulate batch statistics into the metric states."""

    @abstractmethod
    def compute(self) -> Any:
        """Override: compute the final value from the accumulated states."""

    # ------------------------------------------------------------------ misc
    def plot(self, *_: Any, **__: Any) -> Any:
        """Override in subclasses to plot the metric value(s)."""
        raise NotImplementedError

    def _plot(self, val: Optional[Any] = None, ax: Optional[Any] = None) -> Any:
        from metrics_amd.utilities.plot import plot_single_or_multi_val

        val = val if val is not None else self.compute()
        return plot_single_or_multi_val(
            val,
            ax=ax,
            higher_is_better=self.higher_is_better,
            name=self.__class__.__name__,
            lower_bound=self.plot_lower_bound,
            upper_bound=self.plot_upper_bound,
            legend_name=self.plot_legend_name,
        )

    def reset(self) -> None:
        """Reset all metric states to their defaults."""
        if self.__dict__.get("_lazy_dirty"):
            self.__dict__["_lazy_dirty"] = False
            buf = self.__dict__.get("_hip_hist_buf")
            if buf is not None:
                buf.zero_()
        self._update_count = 0
        self._forward_cache = None
        self._computed = None

        for attr, default in self._defaults.items():
            current_val = getattr(self, attr)
            if isinstance(default, Tensor):
                setattr(self, attr, default.detach().clone().to(current_val.device if isinstance(current_val, Tensor) else self.device))
            else:
                setattr(self, attr, [])

        # reset internal sync states
        self._cache = None
        self._is_synced = False
        self._pending_sync_event = None

    def clone(self) -> "Metric":
        """Return a deep copy of the metric."""
        return deepcopy(self)

    def __getstate__(self) -> Dict[str, Any]:
        # the wrapped update/compute closures capture `self` and an inspect
        # signature — drop them here and rebuild in __setstate__
        return {k: v for k, v in self.__dict__.items() if k not in ("update", "compute", "_update_signature")}

    def __setstate__(self, state: Dict[str, Any]) -> None:
        self.__dict__.update(state)
        self._update_signature = inspect.signature(self.update)
        self.update = self._wrap_update(self.update)  # type: ignore[method-assign]
        self.compute = self._wrap_compute(self.compute)  # type: ignore[method-assign]

    # hot internal scalars set on every update/compute: nn.Module.__setattr__
    # walks parameter/buffer/module dicts per call (~1us each, dozens per
    # collection update) — bypass it for names that can never be any of those
    _FAST_ATTRS = frozenset(
        {
            "_computed",
            "_update_count",
            "_forward_cache",
            "_to_sync",
            "_should_unsync",
            "_enable_grad",
            "_is_synced",
            "_pending_sync_event",
        }
    )

    def __setattr__(self, name: str, value: Any) -> None:
        if name in Metric._FAST_ATTRS:
            object.__setattr__(self, name, value)
            return
        if name in ("higher_is_better", "is_differentiable", "full_state_update", "plot_lower_bound", "plot_upper_bound", "plot_legend_name"):
            raise RuntimeError(f"Can't change const `{name}`.")
        # metric STATES are plain tensor/list attributes (never
        # parameters/buffers/modules): skip the nn.Module bookkeeping walk
        defaults = self.__dict__.get("_defaults")
        if defaults is not None and name in defaults and isinstance(value, (Tensor, list)):
            object.__setattr__(self, name, value)
            return
        super().__setattr__(name, value)

    def type(self, dst_type: Union[str, torch.dtype]) -> "Metric":  # noqa: A003
        """Dtype transfers are blocked: use :meth:`set_dtype` instead."""
        return self

    def float(self) -> "Metric":  # noqa: A003
        """Dtype transfers are blocked: use :meth:`set_dtype` instead."""
        return self

    def double(self) -> "Metric":
        """Dtype transfers are blocked: use :meth:`set_dtype` instead."""
        return self

    def half(self) -> "Metric":
        """Dtype transfers are blocked: use :meth:`set_dtype` instead."""
        return self

    def set_dtype(self, dst_type: Union[str, torch.dtype]) -> "Metric":
        """Transfer all metric states to ``dst_type``."""
        self._dtype_convert = True
        out = super().type(dst_type)
        out._dtype_convert = False
        return out

    def _apply(self, fn: Callable, exclude_state: Sequence[str] = ()) -> Module:
        """Extend nn.Module._apply to also move/cast the registered metric states."""
        self._maybe_flush_lazy()
        this = super()._apply(fn)
        fs = str(fn)
        is_dtype_fn = any(
            f in fs for f in ("Module.type", "Module.half", "Module.float", "Module.double", "Module.bfloat16")
        )
        # dtype conversion of states only through set_dtype (which sets the flag)
        if not self._dtype_convert and is_dtype_fn:
            return this

        # also apply fn to metric states and defaults
        for key, value in this._defaults.items():
            if key in exclude_state:
                continue
            if isinstance(value, Tensor):
                this._defaults[key] = fn(value)
            elif isinstance(value, Sequence):
                this._defaults[key] = [fn(v) for v in value]

            current_val = getattr(this, key)
            if isinstance(current_val, Tensor):
                setattr(this, key, fn(current_val))
            elif isinstance(current_val, Sequence):
                setattr(this, key, [fn(cur_v) for cur_v in current_val])
            else:
                raise TypeError(
                    f"Expected metric state to be either a Tensor or a list of Tensor, but encountered {current_val}"
                )

        # refresh the cached device/dtype from a probe tensor
        _dummy = fn(torch.zeros(1, device=this._device))
        this._device = _dummy.device
        this._dtype = _dummy.dtype if _dummy.is_floating_point() else this._dtype

        # additional apply to forward cache and computed attributes (may be nested)
        if this._computed is not None:
            this._computed = apply_to_collection(this._computed, Tensor, fn)
        if this._forward_cache is not None:
            this._forward_cache = apply_to_collection(this._forward_cache, Tensor, fn)
        return this

    def persistent(self, mode: bool = False) -> None:
        """Toggle whether metric states are saved to ``state_dict``."""
        for key in self._persistent:
            self._persistent[key] = mode

    # --------------------------------------------------------- checkpointing
    def state_dict(  # type: ignore[override]
        self,
        destination: Optional[Dict[str, Any]] = None,
        prefix: str = "",
        keep_vars: bool = False,
    ) -> Dict[str, Any]:
        self._maybe_flush_lazy()
        destination = super().state_dict(destination=destination, prefix=prefix, keep_vars=keep_vars)
        # register metric states under `prefix + state_name` (byte-compatible layout)
        for key in self._defaults:
            if not self._persistent[key]:
                continue
            current_val = getattr(self, key)
            if not keep_vars:
                if isinstance(current_val, Tensor):
                    current_val = current_val.detach()
                elif isinstance(current_val, list):
                    current_val = [cur_v.detach() if isinstance(cur_v, Tensor) else cur_v for cur_v in current_val]
            destination[prefix + key] = deepcopy(current_val)
        return destination

    def _load_from_state_dict(
        self,
        state_dict: dict,
        prefix: str,
        local_metadata: dict,
        strict: bool,
        missing_keys: List[str],
        unexpected_keys: List[str],
        error_msgs: List[str],
    ) -> None:
        if self.__dict__.get("_lazy_dirty"):
            self.__dict__["_lazy_dirty"] = False
            buf = self.__dict__.get("_hip_hist_buf")
            if buf is not None:
                buf.zero_()
        for key in self._defaults:
            name = prefix + key
            if name in state_dict:
                setattr(self, key, state_dict.pop(name))
        super()._load_from_state_dict(
            state_dict, prefix, local_metadata, True, missing_keys, unexpected_keys, error_msgs
        )

    def _copy_state_dict(self) -> Dict[str, Union[Tensor, List[Any]]]:
        """Detached deep copy of the current metric states."""
        self._maybe_flush_lazy()
        cache: Dict[str, Union[Tensor, List[Any]]] = {}
        for attr in self._defaults:
            current_value = getattr(self, attr)
            if isinstance(current_value, Tensor):
                cache[attr] = current_value.detach().clone().to(current_value.device)
            else:
                cache[attr] = [
                    _.detach().clone().to(_.device) if isinstance(_, Tensor) else deepcopy(_) for _ in current_value
                ]
        return cache

    def _filter_kwargs(self, **kwargs: Any) -> Dict[str, Any]:
        """Filter kwargs so only those accepted by this metric's ``update`` remain."""
        _params = (inspect.Parameter.VAR_POSITIONAL, inspect.Parameter.VAR_KEYWORD)
        _sign_params = self._update_signature.parameters
        filtered_kwargs = {
            k: v for k, v in kwargs.items() if (k in _sign_params and _sign_params[k].kind not in _params)
        }
        exists_var_keyword = any(v.kind == inspect.Parameter.VAR_KEYWORD for v in _sign_params.values())
        # nothing matched the signature: pass everything through unchanged
        if not filtered_kwargs and not exists_var_keyword:
            filtered_kwargs = kwargs
        if exists_var_keyword:
            filtered_kwargs = kwargs
        return filtered_kwargs

    def __hash__(self) -> int:
        # ID of the state tensors — two metrics only hash equal if they share state
        hash_vals = [self.__class__.__name__]
        for key in self._defaults:
            val = getattr(self, key)
            if isinstance(val, Tensor):
                hash_vals.append(id(val))
            else:
                hash_vals.append(tuple(id(v) for v in val))
        return hash(tuple(hash_vals))

    # ----------------------------------------------------- operator overloads
    def __add__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.add, self, other)

    def __and__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_and, self, other)

    def __eq__(self, other: Any) -> "CompositionalMetric":  # type: ignore[override]
        return CompositionalMetric(torch.eq, self, other)

    def __floordiv__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.floor_divide, self, other)

    def __ge__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.ge, self, other)

    def __gt__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.gt, self, other)

    def __le__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.le, self, other)

    def __lt__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.lt, self, other)

    def __matmul__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.matmul, self, other)

    def __mod__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.fmod, self, other)

    def __mul__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.mul, self, other)

    def __ne__(self, other: Any) -> "CompositionalMetric":  # type: ignore[override]
        return CompositionalMetric(torch.ne, self, other)

    def __or__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_or, self, other)

    def __pow__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.pow, self, other)

    def __radd__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.add, other, self)

    def __rand__(self, other: Any) -> "CompositionalMetric":
        # & is commutative, so keep self first (torch.bitwise_and wants the
        # tensor operand on the left when `other` is a plain int)
        return CompositionalMetric(torch.bitwise_and, self, other)

    def __rfloordiv__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.floor_divide, other, self)

    def __rmatmul__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.matmul, other, self)

    def __rmod__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.fmod, other, self)

    def __rmul__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.mul, other, self)

    def __ror__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_or, other, self)

    def __rpow__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.pow, other, self)

    def __rsub__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.sub, other, self)

    def __rtruediv__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.true_divide, other, self)

    def __rxor__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_xor, other, self)

    def __sub__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.sub, self, other)

    def __truediv__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.true_divide, self, other)

    def __xor__(self, other: Any) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_xor, self, other)

    def __abs__(self) -> "CompositionalMetric":
        return CompositionalMetric(torch.abs, self, None)

    def __inv__(self) -> "CompositionalMetric":
        return CompositionalMetric(torch.bitwise_not, self, None)

    def __invert__(self) -> "CompositionalMetric":
        return self.__inv__()

    def __neg__(self) -> "CompositionalMetric":
        return CompositionalMetric(_neg, self, None)

    def __pos__(self) -> "CompositionalMetric":
        return CompositionalMetric(torch.abs, self, None)

    def __getitem__(self, idx: Any) -> "CompositionalMetric":
        return CompositionalMetric(lambda x: x[idx], self, None)

    def __getnewargs__(self) -> tuple:
        return tuple(Metric.__str__(self))

    __iter__ = None  # type: ignore[assignment]


def _neg(x: Tensor) -> Tensor:
    return -torch.abs(x)


class CompositionalMetric(Metric):
    """Lazy elementwise composition over the compute() results of two metrics."""

    def __init__(
        self,
        operator: Callable,
        metric_a: Union[Metric, float, Tensor],
        metric_b: Union[Metric, float, Tensor, None],
    ) -> None:
        super().__init__()

        self.op = operator

        if isinstance(metric_a, Tensor):
            self.register_buffer("metric_a", metric_a, persistent=False)
        else:
            self.metric_a = metric_a

        if isinstance(metric_b, Tensor):
            self.register_buffer("metric_b", metric_b, persistent=False)
        else:
            self.metric_b = metric_b

    def _sync_dist(self, dist_sync_fn: Optional[Callable] = None, process_group: Optional[Any] = None) -> None:
        # No syncing required here: children sync themselves.
        pass

    def update(self, *args: Any, **kwargs: Any) -> None:
        if isinstance(self.metric_a, Metric):
            self.metric_a.update(*args, **self.metric_a._filter_kwargs(**kwargs))
        if isinstance(self.metric_b, Metric):
            self.metric_b.update(*args, **self.metric_b._filter_kwargs(**kwargs))

    def compute(self) -> Any:
        val_a = self.metric_a.compute() if isinstance(self.metric_a, Metric) else self.metric_a
        val_b = self.metric_b.compute() if isinstance(self.metric_b, Metric) else self.metric_b

        if val_b is None:
            return self.op(val_a)
        return self.op(val_a, val_b)

    @torch.jit.unused
    def forward(self, *args: Any, **kwargs: Any) -> Any:
        val_a = (
            self.metric_a(*args, **self.metric_a._filter_kwargs(**kwargs))
            if isinstance(self.metric_a, Metric)
            else self.metric_a
        )
        val_b = (
            self.metric_b(*args, **self.metric_b._filter_kwargs(**kwargs))
            if isinstance(self.metric_b, Metric)
            else self.metric_b
        )

        if val_a is None:
            self._forward_cache = None
            return self._forward_cache

        if val_b is None:
            if isinstance(self.metric_b, Metric):
                self._forward_cache = None
                return self._forward_cache
            # Unary op
            self._forward_cache = self.op(val_a)
            return self._forward_cache

        # Binary op
        self._forward_cache = self.op(val_a, val_b)
        return self._forward_cache

    def reset(self) -> None:
        if isinstance(self.metric_a, Metric):
            self.metric_a.reset()
        if isinstance(self.metric_b, Metric):
            self.metric_b.reset()

    def persistent(self, mode: bool = False) -> None:
        if isinstance(self.metric_a, Metric):
            self.metric_a.persistent(mode=mode)
        if isinstance(self.metric_b, Metric):
            self.metric_b.persistent(mode=mode)

    def __repr__(self) -> str:
        _op_metrics = f"(\n  {self.op.__name__}(\n    {self.metric_a!r},\n    {self.metric_b!r}\n  )\n)"
        return self.__class__.__name__ + _op_metrics

    def _wrap_compute(self, compute: Callable) -> Callable:
        return compute
