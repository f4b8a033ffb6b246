"""L1 — distributed state synchronization over RCCL/xGMI.

Parity target: torchmetrics ``utilities/distributed.py`` (gather_all_tensors)
plus ``Metric._sync_dist`` semantics (metric.py:501-540).

MI355X-native design (differs from the reference on purpose):

The reference syncs every state with an all-gather followed by a *local*
reduction, because its sync API is gather-shaped. On an MI355X node the 8 GPUs
are connected point-to-point over xGMI (7 links x ~153 GB/s), so collective
latency — not bandwidth — dominates for the tiny fixed-shape states metrics
carry (scalars, (C,), (C,C), (T,2,2)). We therefore:

1. map ``sum``/``mean``/``max``/``min`` reductions onto true RCCL
   ``all_reduce`` (one collective, no world_size-times memory),
2. *fuse* all same-dtype all-reducible states of a metric into one flat
   buffer => ONE RCCL launch per (dtype, op) instead of one per state,
3. issue every collective with ``async_op=True`` and wait once at the end, so
   the per-collective launch latencies overlap on the RCCL stream,
4. keep the gather path (shape exchange + pad/trim) only for ``cat`` / ``None``
   / user-callable reductions, where rank order must be preserved.

Autograd: a state that ``requires_grad`` always takes the gather path, where
the local rank's slot is replaced by the autograd-connected local tensor
(same contract as the reference).
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple, Union

import torch
import torch.distributed as dist
from torch import Tensor


def reduce(x: Tensor, reduction: str) -> Tensor:
    """Reduce a tensor according to ``reduction`` in {'elementwise_mean','sum','none'}."""
    if reduction == "elementwise_mean":
        return torch.mean(x)
    if reduction == "none" or reduction is None:
        return x
    if reduction == "sum":
        return torch.sum(x)
    raise ValueError("Reduction parameter unknown.")


def class_reduce(num: Tensor, denom: Tensor, weights: Tensor, class_reduction: str = "none") -> Tensor:
    """Reduce per-class fractions num/denom with micro/macro/weighted/none semantics."""
    valid_reduction = ("micro", "macro", "weighted", "none", None)
    fraction = torch.sum(num) / torch.sum(denom) if class_reduction == "micro" else num / denom
    fraction[fraction != fraction] = 0  # nan from 0/0
    if class_reduction == "micro":
        return fraction
    if class_reduction == "macro":
        return torch.mean(fraction)
    if class_reduction == "weighted":
        return torch.sum(fraction * (weights.float() / torch.sum(weights)))
    if class_reduction == "none" or class_reduction is None:
        return fraction
    raise ValueError(f"Reduction parameter {class_reduction} unknown, choose one of {valid_reduction}")


def _simple_gather_all_tensors(result: Tensor, group: Any, world_size: int) -> List[Tensor]:
    gathered_result = [torch.zeros_like(result) for _ in range(world_size)]
    dist.all_gather(gathered_result, result, group)
    # keep the local slot connected to the autograd graph
    gathered_result[dist.get_rank(group)] = result
    return gathered_result


def gather_all_tensors(result: Tensor, group: Optional[Any] = None) -> List[Tensor]:
    """All-gather a tensor across the process group, handling uneven shapes.

    Returns a list of ``world_size`` tensors in rank order; the local rank's
    entry is the input tensor itself (autograd-connected).
    """
    if group is None:
        group = dist.group.WORLD
    # collectives need contiguous buffers
    result = result.contiguous()
    world_size = dist.get_world_size(group)
    if not _use_side_stream(group):
        # the reference barriers before every gather; on RCCL the collectives
        # are stream-ordered and all_gather is itself a synchronization point,
        # so the barrier is a pure extra collective — keep it only for
        # host-blocking backends (gloo), where it preserves reference timing
        dist.barrier(group=group)

    if result.ndim == 0:
        return _simple_gather_all_tensors(result, group, world_size)

    # 1. exchange shapes
    local_size = torch.tensor(result.shape, device=result.device)
    local_sizes = [torch.zeros_like(local_size) for _ in range(world_size)]
    dist.all_gather(local_sizes, local_size, group=group)
    max_size = torch.stack(local_sizes).max(dim=0).values
    all_sizes_equal = all(all(ls == max_size) for ls in local_sizes)
    if all_sizes_equal:
        return _simple_gather_all_tensors(result, group, world_size)

    # 2. pad to the max shape, gather, trim back per-rank
    pad_dims = []
    pad_by = (max_size - local_size).detach().cpu()
    for val in reversed(pad_by):
        pad_dims.append(0)
        pad_dims.append(val.item())
    result_padded = torch.nn.functional.pad(result, pad_dims)
    gathered_result = [torch.zeros_like(result_padded) for _ in range(world_size)]
    dist.all_gather(gathered_result, result_padded, group)
    for idx, item_size in enumerate(local_sizes):
        slice_param = [slice(dim_size) for dim_size in item_size]
        gathered_result[idx] = gathered_result[idx][slice_param]
    gathered_result[dist.get_rank(group)] = result
    return gathered_result


# ---------------------------------------------------------------------------
# RCCL fast-path sync engine
# ---------------------------------------------------------------------------

_ALLREDUCE_OPS = {"sum": "sum", "mean": "sum", "max": "max", "min": "min"}

# One dedicated HIP stream per device for state-sync collectives, so RCCL
# kernels overlap with the next update()'s compute kernels on the default
# stream (BASELINE north-star: side-stream overlapped sync).
_SYNC_STREAMS: Dict[int, "torch.cuda.Stream"] = {}


def _sync_stream(device: torch.device) -> "torch.cuda.Stream":
    s = _SYNC_STREAMS.get(device.index)
    if s is None:
        s = torch.cuda.Stream(device=device)
        _SYNC_STREAMS[device.index] = s
    return s


def _use_side_stream(group: Any) -> bool:
    """Side-stream sync only makes sense for stream-ordered (RCCL) backends."""
    if not torch.cuda.is_available():
        return False
    try:
        return "nccl" in str(dist.get_backend(group)).lower()
    except RuntimeError:
        return False


def _reduction_kind(reduce_fn: Union[str, Callable, None]) -> str:
    """Classify a dist_reduce_fx (already canonicalized to a string or callable)."""
    if isinstance(reduce_fn, str):
        return reduce_fn
    if reduce_fn is None:
        return "none"
    return "custom"


def sync_states_fast(
    states: Dict[str, Union[Tensor, List[Tensor]]],
    kinds: Dict[str, str],
    custom_fns: Dict[str, Callable],
    group: Optional[Any] = None,
    gather_fn: Optional[Callable] = None,
    overlap: bool = True,
) -> Tuple[Dict[str, Union[Tensor, List[Tensor]]], Optional["torch.cuda.Event"]]:
    """Synchronize a metric's states across the process group.

    ``kinds[name]`` in {'sum','mean','max','min','cat','none','custom'};
    ``custom_fns[name]`` holds the callable for kind == 'custom'.

    ``gather_fn``: if given (a user-supplied dist_sync_fn), EVERY state takes
    the gather path through it — reference semantics for custom sync fns.

    Returns ``(synced, done_event)``. List states come back as a single
    concatenated tensor for 'cat' (reference behavior) and as rank-order
    flattened lists otherwise.

    Overlap: with a RCCL backend the fused all-reduce path is issued from a
    dedicated side HIP stream. Only the cheap state *read* (flat-buffer build)
    is fenced back onto the caller's stream, so the next ``update()`` launches
    immediately while the collectives run; ``done_event`` (non-None iff the
    side stream was used) must be waited on by whatever stream *reads* the
    synced values — :meth:`Metric._wait_pending_sync` does this lazily in
    ``compute``.
    """
    if group is None:
        group = dist.group.WORLD
    world_size = dist.get_world_size(group)
    out: Dict[str, Any] = {}

    # --- partition states -------------------------------------------------
    fuse_buckets: Dict[Tuple[torch.dtype, torch.device, str], List[Tuple[str, Tensor]]] = {}
    gather_names: List[str] = []
    for name, val in states.items():
        kind = kinds[name]
        if (
            gather_fn is None
            and isinstance(val, Tensor)
            and kind in _ALLREDUCE_OPS
            and not val.requires_grad
        ):
            op = _ALLREDUCE_OPS[kind]
            dt = val.dtype if val.is_floating_point() else torch.long
            fuse_buckets.setdefault((dt, val.device, op), []).append((name, val))
        else:
            gather_names.append(name)

    works = []
    _gf = gather_fn if gather_fn is not None else gather_all_tensors

    side = None
    done_event = None
    if fuse_buckets and overlap and _use_side_stream(group):
        dev = next(iter(fuse_buckets))[1]
        if dev.type == "cuda":
            side = _sync_stream(dev)

    # --- fused all-reduce buckets ----------------------------------------
    red_op = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX, "min": dist.ReduceOp.MIN}
    bucket_views: List[Tuple[Tensor, List[Tuple[str, Tensor]], str]] = []

    def _build_buffers() -> None:
        # the only part that READS the state tensors — the caller's stream
        # need only be fenced behind this, not behind the collectives
        for (dt, _dev, op), entries in fuse_buckets.items():
            if len(entries) == 1:
                name, val = entries[0]
                buf = val.contiguous().to(dt) if val.dtype != dt else val.contiguous().clone()
            else:
                buf = torch.cat([v.reshape(-1).to(dt) for _, v in entries])
            bucket_views.append((buf, entries, op))

    def _issue_allreduces() -> None:
        for buf, _entries, op in bucket_views:
            works.append(dist.all_reduce(buf, op=red_op[op], group=group, async_op=True))

    def _finish_allreduces() -> None:
        for w in works:
            if w is not None:
                w.wait()
        for buf, entries, _op in bucket_views:
            if len(entries) == 1:
                name, val = entries[0]
                res = buf.view(val.shape)
                if kinds[name] == "mean":
                    res = res / world_size
                out[name] = res.to(val.dtype) if res.dtype != val.dtype and kinds[name] != "mean" else res
            else:
                offset = 0
                for name, val in entries:
                    n = val.numel()
                    res = buf[offset : offset + n].view(val.shape)
                    offset += n
                    if kinds[name] == "mean":
                        res = res / world_size
                    elif res.dtype != val.dtype:
                        res = res.to(val.dtype)
                    out[name] = res

    if side is not None:
        main_stream = torch.cuda.current_stream()
        produced = torch.cuda.Event()
        produced.record(main_stream)
        with torch.cuda.stream(side):
            # order after the kernels that produced the current state values
            side.wait_event(produced)
            _build_buffers()
            # the states have now been read into the flat buffers
            read_done = torch.cuda.Event()
            read_done.record(side)
            _issue_allreduces()
            _finish_allreduces()  # stream-level waits + scatter-back views
            done_event = torch.cuda.Event()
            done_event.record(side)
        # later writers of the original state tensors (the next update) must
        # not overwrite them before the side stream has copied them out —
        # this only costs the D2D copy time, not the collective time
        main_stream.wait_event(read_done)

    if side is None:
        # inline path: issue the async all-reduces first so their launch
        # latencies overlap with the gather collectives below
        _build_buffers()
        _issue_allreduces()

    # --- gather path (cat / none / custom / autograd) ---------------------
    gathered: Dict[str, Any] = {}
    for name in gather_names:
        val = states[name]
        kind = kinds[name]
        if isinstance(val, list):
            if len(val) == 0:
                # other ranks may still hold data: contribute a zero-size
                # tensor so the uneven-shape gather proceeds
                placeholder = torch.zeros(0, device=_any_device(states))
                gathered[name] = ("list_cat", _gf(placeholder, group))
            elif kind == "cat":
                # pre-concat list states into one tensor: one gather per state
                gathered[name] = ("list_cat", _gf(dim_zero_cat_local(val), group))
            else:
                # none / custom list states gather each element separately and
                # flatten in (element, rank) order — reference semantics
                gathered[name] = ("list_elems", [_gf(e, group) for e in val])
        else:
            gathered[name] = ("tensor", _gf(val, group))

    if side is None:
        _finish_allreduces()

    # --- finalize gather states: apply the reference's post-gather reduction
    for name, (tag, gath) in gathered.items():
        kind = kinds[name]
        if tag == "list_elems":
            flat = [t for per_elem in gath for t in per_elem]  # (element, rank) order
            if kind == "custom":
                out[name] = custom_fns[name](flat)
            else:  # 'none'
                out[name] = flat
            continue
        if tag == "list_cat":
            nonempty = [t for t in gath if t.numel() > 0]
            if not nonempty:
                # every rank was empty: the synced state is a zero-size tensor
                # (reference behavior — dim_zero_cat of the placeholder gathers),
                # so downstream compute sees an empty tensor, not an empty list
                out[name] = torch.cat(gath, dim=0) if gath else []
            else:
                out[name] = torch.cat(nonempty, dim=0)
            continue
        # tag == 'tensor'
        shapes_equal = all(t.shape == gath[0].shape for t in gath)
        if kind == "cat":
            # reference stacks equal-shape tensor states then applies
            # dim_zero_cat (identity on a Tensor) => a (world, ...) stack
            out[name] = torch.stack(gath, dim=0) if shapes_equal else torch.cat(gath, dim=0)
        elif kind in ("sum", "mean", "max", "min"):
            stacked = torch.stack(gath, dim=0)
            if kind == "sum":
                out[name] = stacked.sum(0)
            elif kind == "mean":
                out[name] = stacked.float().mean(0)
            elif kind == "max":
                out[name] = stacked.max(0).values
            else:
                out[name] = stacked.min(0).values
        elif kind == "none":
            out[name] = torch.stack(gath, dim=0) if shapes_equal else gath
        else:  # custom callable
            stacked = torch.stack(gath, dim=0) if shapes_equal else gath
            out[name] = custom_fns[name](stacked)
    return out, done_event


def _any_device(states: Dict[str, Any]) -> torch.device:
    for v in states.values():
        if isinstance(v, Tensor):
            return v.device
        if isinstance(v, list) and v and isinstance(v[0], Tensor):
            return v[0].device
    return torch.device("cpu")


def dim_zero_cat_local(x: List[Tensor]) -> Tensor:
    x = [y.unsqueeze(0) if y.numel() == 1 and y.ndim == 0 else y for y in x]
    return torch.cat(x, dim=0)
