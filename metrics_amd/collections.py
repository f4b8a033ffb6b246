"""L5 — MetricCollection with compute-group state dedup.

Parity: torchmetrics ``collections.py`` (MetricCollection semantics:
dict/list/args inputs, prefix/postfix renaming, nested-collection flattening,
per-metric kwarg routing, compute groups with shared-by-reference state,
copy-on-access).

Compute groups: metrics whose states are equal after the first update (e.g.
Accuracy/Precision/Recall/F1, which all accumulate tp/fp/tn/fn) are merged
into one group; subsequent ``update`` calls only run the group leader's
update and members alias the leader's state tensors. Accessing metrics via
``items()/values()/[key]`` deep-copies states to break aliasing (reference is
re-established on the next update). On an MI355X node this is the difference
between one fused stat-scores kernel launch per batch and N of them.
"""
from __future__ import annotations

from collections import OrderedDict
from copy import deepcopy
from typing import Any, Dict, Hashable, Iterable, Iterator, List, Mapping, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor
from torch.nn import ModuleDict

from metrics_amd.metric import Metric
from metrics_amd.utilities import tracing
from metrics_amd.utilities.data import _flatten_dict, allclose
from metrics_amd.utilities.prints import rank_zero_warn


def _strip_prefix(s: str, prefix: str) -> str:
    return s[len(prefix):] if s.startswith(prefix) else s


def _strip_suffix(s: str, suffix: str) -> str:
    return s[: -len(suffix)] if s.endswith(suffix) else s


class _FusedMulticlassUpdatePlan:
    """Collection-level cross-metric fusion: the stat-scores, confusion-matrix
    and exact-match group leaders all start with the same argmax pass over the
    (B, C) logits — ONE fused kernel call feeds all of them (MI355X-first:
    the logits are read once from HBM instead of three times)."""

    def __init__(self, stat, confmat, exact, curve):
        self.stat = stat
        self.confmat = confmat
        self.exact = exact
        self.curve = curve
        self.leaders = tuple(m for m in (stat, confmat, exact, curve) if m is not None)

    @staticmethod
    def build(collection) -> "Optional[_FusedMulticlassUpdatePlan]":
        stat = confmat = exact = curve = None
        for members in collection._groups.values():
            leader = getattr(collection, members[0])
            kind = getattr(type(leader), "_hip_fused_kind", None)
            if kind is None or getattr(leader, "validate_args", True):
                continue  # validations are skipped on the fused path
            if kind == "mc_stat" and stat is None:
                if getattr(leader, "multidim_average", "global") == "global" and getattr(leader, "top_k", 1) == 1:
                    stat = leader
            elif kind == "mc_confmat" and confmat is None:
                confmat = leader
            elif kind == "mc_exact" and exact is None:
                if getattr(leader, "multidim_average", "global") == "global":
                    exact = leader
            elif kind == "mc_curve" and curve is None:
                import os

                if (
                    leader.thresholds is not None
                    and getattr(leader, "average", None) != "micro"
                    and os.environ.get("METRICS_AMD_FUSE_CURVE", "1") != "0"
                ):
                    curve = leader
        participants = [m for m in (stat, confmat, exact, curve) if m is not None]
        if stat is None or len(participants) < 2:
            return None
        ncs = {m.num_classes for m in participants}
        igs = {m.ignore_index for m in participants}
        if len(ncs) != 1 or len(igs) != 1:
            return None
        return _FusedMulticlassUpdatePlan(stat, confmat, exact, curve)

    def try_run(self, *args: Any, **kwargs: Any) -> tuple:
        """Run the fused update if the inputs qualify; returns the handled
        leaders (empty tuple -> caller falls back to per-leader updates)."""
        if kwargs or len(args) != 2:
            return ()
        preds, target = args
        if not (isinstance(preds, Tensor) and isinstance(target, Tensor) and preds.is_cuda):
            return ()
        if target.ndim != 1:
            return ()
        if preds.is_floating_point():
            if preds.ndim != 2 or preds.dtype not in (torch.float32, torch.bfloat16):
                return ()
        elif preds.shape != target.shape:
            return ()
        from metrics_amd.ops import _hip

        if not _hip.hip_available():
            return ()
        stat = self.stat
        scratch = getattr(stat, "_hip_scratch", None)
        if scratch is None or scratch.device != preds.device:
            scratch = torch.zeros(3 * stat.num_classes + 1, dtype=torch.long, device=preds.device)
            stat._hip_scratch = scratch
        curve = self.curve
        rowstats = None
        if curve is not None and preds.is_floating_point():
            # the fused pass also emits softmax row-stats + the epoch flag;
            # the curve leader's own row-stats kernel is skipped
            rbuf = curve.__dict__.get("_hip_rowstats_buf")
            if rbuf is None or rbuf.device != preds.device or rbuf.shape[1] < preds.shape[0]:
                rbuf = torch.empty(2, preds.shape[0], dtype=torch.float32, device=preds.device)
                curve.__dict__["_hip_rowstats_buf"] = rbuf
            ebuf = _hip._epoch_buf(preds.device, curve)
            rowstats = (rbuf[0], rbuf[1], ebuf)
        elif curve is not None:
            curve = None  # label preds: curve can't ride the fused pass
        deferred = _hip.mc_fused_collection_update(
            preds, target, stat.num_classes, stat.ignore_index,
            stat=(scratch, stat.tp, stat.fp, stat.tn, stat.fn),
            confmat=self.confmat.confmat if self.confmat is not None else None,
            exact=(self.exact.correct, self.exact.total) if self.exact is not None else None,
            rowstats=rowstats,
            # the curve hist must read the epoch flag BEFORE it closes: defer
            # the apply launch behind the hist so its single block closes the
            # epoch for free (saves a dedicated ~4us bump dispatch per step)
            defer_apply=curve is not None,
            bump_epoch_ptr=rowstats[2].data_ptr() if curve is not None else 0,
        )
        if curve is not None:
            _hip.curve_hist_into_confmat(
                preds, target, curve.thresholds, curve.ignore_index, curve.confmat,
                mode=0, norm="softmax", owner=curve, stats_ready=True, skip_epoch_bump=True,
            )
            deferred()
            handled = self.leaders
        else:
            handled = tuple(m for m in self.leaders if m is not self.curve)
        return handled


class MetricCollection(ModuleDict):
    """A dict-like container of metrics sharing one ``update``/``compute`` call pattern.

    Args:
        metrics: a single Metric, a sequence of Metrics (keyed by class name),
            or a dict name->Metric (keys sorted alphabetically). Nested
            ``MetricCollection`` inputs are flattened.
        additional_metrics: further metrics when ``metrics`` is not a dict.
        prefix / postfix: strings pre/appended to every output key.
        compute_groups: True (auto-detect groups after the first update),
            False (disable), or an explicit list of lists of metric names.
    """

    _modules: Dict[str, Metric]  # type: ignore[assignment]

    def __init__(
        self,
        metrics: Union[Metric, Sequence[Metric], Dict[str, Metric]],
        *additional_metrics: Metric,
        prefix: Optional[str] = None,
        postfix: Optional[str] = None,
        compute_groups: Union[bool, List[List[str]]] = True,
    ) -> None:
        super().__init__()
        self._fused_plan: Any = False
        self.prefix = self._check_arg(prefix, "prefix")
        self.postfix = self._check_arg(postfix, "postfix")
        self._enable_compute_groups = compute_groups
        self._groups_checked: bool = False
        self._state_is_copy: bool = False

        self.add_metrics(metrics, *additional_metrics)

    # ------------------------------------------------------------------ calls
    @property
    def metric_state(self) -> Dict[str, Dict[str, Any]]:
        """States of every metric in the collection."""
        return {k: m.metric_state for k, m in self.items(keep_base=False, copy_state=False)}

    @torch.jit.unused
    def forward(self, *args: Any, **kwargs: Any) -> Dict[str, Any]:
        """Call forward on every metric; kwargs are routed per-metric by update signature."""
        return self._compute_and_reduce("forward", *args, **kwargs)

    def update(self, *args: Any, **kwargs: Any) -> None:
        """Call update on every metric (or only group leaders once groups are formed)."""
        if self._groups_checked:
            # invalidate cached compute results on ALL members
            for k in self.keys(keep_base=True):
                getattr(self, str(k))._computed = None
            if self._fused_plan is False:
                self._fused_plan = _FusedMulticlassUpdatePlan.build(self)
            fused_done: tuple = ()
            if self._fused_plan is not None:
                with tracing.range("MetricCollection.fused_update"):
                    fused_done = self._fused_plan.try_run(*args, **kwargs)
            # run only each group's leader
            for members in self._groups.values():
                leader = getattr(self, members[0])
                # identity check: Metric.__eq__ is the compositional operator
                # (returns a CompositionalMetric, which is always truthy)
                if any(leader is m for m in fused_done):
                    leader._update_count += 1
                    continue
                leader.update(*args, **leader._filter_kwargs(**kwargs))
            if self._state_is_copy:
                # re-establish the alias links broken by a copy-on-access
                self._compute_groups_create_state_ref()
                self._state_is_copy = False
            return

        # first update: run every metric, then detect groups from equal states
        for m in self.values(copy_state=False):
            m.update(*args, **m._filter_kwargs(**kwargs))
        if self._enable_compute_groups:
            # group detection compares state VALUES: lazily-accumulated curve
            # histograms would leave every curve confmat at zero and merge
            # unrelated metrics — materialize first
            for m in self.values(copy_state=False):
                m._maybe_flush_lazy()
            self._merge_compute_groups()
            self._compute_groups_create_state_ref()
            self._groups_checked = True

    def compute(self) -> Dict[str, Any]:
        """Compute the result of every metric; one state sync per compute group."""
        return self._compute_and_reduce("compute")

    def _compute_and_reduce(self, method_name: str, *args: Any, **kwargs: Any) -> Dict[str, Any]:
        # lazy curve histograms live on group LEADERS; members alias the
        # leader's states and may compute first — materialize before iterating
        if self._groups_checked:
            for members in self._groups.values():
                getattr(self, members[0])._maybe_flush_lazy()
        result = {}
        for k, m in self.items(keep_base=True, copy_state=False):
            if method_name == "compute":
                result[k] = m.compute()
            elif method_name == "forward":
                result[k] = m(*args, **m._filter_kwargs(**kwargs))
            else:
                raise ValueError(f"method_name should be either 'compute' or 'forward', but got {method_name}")

        _, duplicates = _flatten_dict(result)

        flat: Dict[str, Any] = {}
        for k, m in self.items(keep_base=True, copy_state=False):
            res = result[k]
            if isinstance(res, dict):
                for key, v in res.items():
                    if duplicates:
                        base_k = _strip_prefix(k, getattr(m, "prefix", None) or "")
                        base_k = _strip_suffix(base_k, getattr(m, "postfix", None) or "")
                        key = f"{base_k}_{key}"
                    if getattr(m, "_from_collection", None) and m.prefix is not None:
                        key = f"{m.prefix}{key}"
                    if getattr(m, "_from_collection", None) and m.postfix is not None:
                        key = f"{key}{m.postfix}"
                    flat[key] = v
            else:
                flat[k] = res
        return {self._set_name(k): v for k, v in flat.items()}

    def reset(self) -> None:
        """Reset every metric."""
        for m in self.values(copy_state=False):
            m.reset()
        if self._enable_compute_groups and self._groups_checked:
            # resets allocate fresh default tensors: re-link group members
            self._compute_groups_create_state_ref()

    # ---------------------------------------------------------- compute groups
    def _merge_compute_groups(self) -> None:
        """Merge groups whose (leader) states compare equal, until a fixed point."""
        merged = True
        while merged:
            merged = False
            ids = list(self._groups.keys())
            for i in ids:
                if i not in self._groups:
                    continue
                for j in ids:
                    if j == i or j not in self._groups:
                        continue
                    m_i = getattr(self, self._groups[i][0])
                    m_j = getattr(self, self._groups[j][0])
                    if self._equal_metric_states(m_i, m_j):
                        self._groups[i].extend(self._groups.pop(j))
                        merged = True
        # re-index 0..n-1
        self._groups = {idx: members for idx, members in enumerate(self._groups.values())}

    @staticmethod
    def _equal_metric_states(metric1: Metric, metric2: Metric) -> bool:
        """True if two metrics hold state that is equal in keys, shapes and values."""
        if len(metric1._defaults) == 0 or len(metric2._defaults) == 0:
            return False
        if metric1._defaults.keys() != metric2._defaults.keys():
            return False
        for key in metric1._defaults:
            s1, s2 = getattr(metric1, key), getattr(metric2, key)
            if type(s1) is not type(s2):
                return False
            if isinstance(s1, Tensor):
                if s1.shape != s2.shape or not allclose(s1, s2):
                    return False
            elif isinstance(s1, list):
                if len(s1) != len(s2):
                    return False
                if not all(a.shape == b.shape and allclose(a, b) for a, b in zip(s1, s2)):
                    return False
        return True

    def _compute_groups_create_state_ref(self, copy: bool = False) -> None:
        """Alias (or deep-copy) the leader's states onto every group member."""
        if copy:
            for members in self._groups.values():
                getattr(self, members[0])._maybe_flush_lazy()
        if not self._state_is_copy:
            for members in self._groups.values():
                leader = getattr(self, members[0])
                for name in members[1:]:
                    member = getattr(self, name)
                    for state in leader._defaults:
                        leader_state = getattr(leader, state)
                        setattr(member, state, deepcopy(leader_state) if copy else leader_state)
                    member._update_count = deepcopy(leader._update_count) if copy else leader._update_count
        self._state_is_copy = copy

    @property
    def compute_groups(self) -> Dict[int, List[str]]:
        """The current compute groups (group index -> member names)."""
        return self._groups

    # ------------------------------------------------------------- management
    def add_metrics(
        self, metrics: Union[Metric, Sequence[Metric], Dict[str, Metric]], *additional_metrics: Metric
    ) -> None:
        """Add metrics to the collection."""
        if isinstance(metrics, Metric):
            metrics = [metrics]
        if isinstance(metrics, Sequence):
            metrics = list(metrics)
            remain: list = []
            for m in additional_metrics:
                (metrics if isinstance(m, (Metric, MetricCollection)) else remain).append(m)
            if remain:
                rank_zero_warn(
                    f"You have passed extra arguments {remain} which are not `Metric` so they will be ignored."
                )
        elif additional_metrics:
            raise ValueError(
                f"You have passed extra arguments {additional_metrics} which are not compatible"
                f" with first passed dictionary {metrics} so they will be ignored."
            )

        if isinstance(metrics, dict):
            for name in sorted(metrics.keys()):
                metric = metrics[name]
                if not isinstance(metric, (Metric, MetricCollection)):
                    raise ValueError(
                        f"Value {metric} belonging to key {name} is not an instance of"
                        " `metrics_amd.Metric` or `metrics_amd.MetricCollection`"
                    )
                if isinstance(metric, Metric):
                    self[name] = metric
                else:  # flatten a nested collection, carrying its renaming
                    for k, v in metric.items(keep_base=False):
                        v.postfix = metric.postfix
                        v.prefix = metric.prefix
                        v._from_collection = True
                        self[f"{name}_{k}"] = v
        elif isinstance(metrics, Sequence):
            for metric in metrics:
                if not isinstance(metric, (Metric, MetricCollection)):
                    raise ValueError(
                        f"Input {metric} to `MetricCollection` is not a instance of"
                        " `metrics_amd.Metric` or `metrics_amd.MetricCollection`"
                    )
                if isinstance(metric, Metric):
                    name = metric.__class__.__name__
                    if name in self:
                        raise ValueError(f"Encountered two metrics both named {name}")
                    self[name] = metric
                else:
                    for k, v in metric.items(keep_base=False):
                        v.postfix = metric.postfix
                        v.prefix = metric.prefix
                        v._from_collection = True
                        self[k] = v
        else:
            raise ValueError(
                "Unknown input to MetricCollection. Expected `Metric`, `MetricCollection` or `dict`/`sequence` of the"
                f" previous, but got {metrics}"
            )

        self._groups_checked = False
        if self._enable_compute_groups:
            self._init_compute_groups()
        else:
            self._groups = {}

    def _init_compute_groups(self) -> None:
        if isinstance(self._enable_compute_groups, list):
            self._groups = dict(enumerate(self._enable_compute_groups))
            for members in self._groups.values():
                for metric in members:
                    if metric not in self:
                        raise ValueError(
                            f"Input {metric} in `compute_groups` argument does not match a metric in the collection."
                            f" Please make sure that {self._enable_compute_groups} matches {self.keys(keep_base=True)}"
                        )
            self._groups_checked = True
        else:
            self._groups = {i: [str(k)] for i, k in enumerate(self.keys(keep_base=True))}

    def clone(self, prefix: Optional[str] = None, postfix: Optional[str] = None) -> "MetricCollection":
        """Deep copy of the collection, optionally re-keyed."""
        mc = deepcopy(self)
        if prefix:
            mc.prefix = self._check_arg(prefix, "prefix")
        if postfix:
            mc.postfix = self._check_arg(postfix, "postfix")
        return mc

    def persistent(self, mode: bool = True) -> None:
        """Toggle state persistence on every metric."""
        for m in self.values(copy_state=False):
            m.persistent(mode)

    def set_dtype(self, dst_type: Union[str, torch.dtype]) -> "MetricCollection":
        """Cast the states of every metric."""
        for m in self.values(copy_state=False):
            m.set_dtype(dst_type)
        return self

    # ------------------------------------------------------------- dict-style
    def _set_name(self, base: str) -> str:
        name = base if self.prefix is None else self.prefix + base
        return name if self.postfix is None else name + self.postfix

    def _to_renamed_dict(self) -> Mapping[str, Metric]:
        out = OrderedDict()
        for k, v in self._modules.items():
            out[self._set_name(k)] = v
        return out

    def __iter__(self) -> Iterator[Hashable]:
        return iter(self.keys())

    def keys(self, keep_base: bool = False) -> Iterable[Hashable]:
        """Keys, with prefix/postfix applied unless ``keep_base``."""
        if keep_base:
            return self._modules.keys()
        return self._to_renamed_dict().keys()

    def items(self, keep_base: bool = False, copy_state: bool = True) -> Iterable[Tuple[str, Metric]]:
        """(key, metric) pairs; by default breaks group-state aliasing via copies."""
        self._compute_groups_create_state_ref(copy_state)
        if keep_base:
            return self._modules.items()
        return self._to_renamed_dict().items()

    def values(self, copy_state: bool = True) -> Iterable[Metric]:
        """Metrics; by default breaks group-state aliasing via copies."""
        self._compute_groups_create_state_ref(copy_state)
        return self._modules.values()

    def __getitem__(self, key: str, copy_state: bool = True) -> Metric:
        """Look up a single metric (prefix/postfix-stripped key)."""
        self._compute_groups_create_state_ref(copy_state)
        if self.prefix:
            key = _strip_prefix(key, self.prefix)
        if self.postfix:
            key = _strip_suffix(key, self.postfix)
        return self._modules[key]

    @staticmethod
    def _check_arg(arg: Optional[str], name: str) -> Optional[str]:
        if arg is None or isinstance(arg, str):
            return arg
        raise ValueError(f"Expected input `{name}` to be a string, but got {type(arg)}")

    def __repr__(self) -> str:
        repr_str = super().__repr__()[:-2]
        if self.prefix:
            repr_str += f",\n  prefix={self.prefix}{',' if self.postfix else ''}"
        if self.postfix:
            repr_str += f"{',' if not self.prefix else ''}\n  postfix={self.postfix}"
        return repr_str + "\n)"

    def plot(
        self,
        val: Optional[Union[dict, Sequence[dict]]] = None,
        ax: Optional[Any] = None,
        together: bool = False,
    ) -> Sequence[Any]:
        """Plot all metric values — one axis per metric, or all in one if ``together``."""
        from metrics_amd.utilities.plot import plot_single_or_multi_val

        if not isinstance(together, bool):
            raise ValueError(f"Expected argument `together` to be a boolean, but got {type(together)}")
        val = val or self.compute()
        if together:
            return plot_single_or_multi_val(val, ax=ax)
        fig_axs = []
        for i, (k, m) in enumerate(self.items(keep_base=False, copy_state=False)):
            if isinstance(val, dict):
                f, a = m.plot(val[k], ax=ax[i] if ax is not None else ax)
            elif isinstance(val, Sequence):
                f, a = m.plot([v[k] for v in val], ax=ax[i] if ax is not None else ax)
            else:
                raise ValueError(f"Unsupported value type for plotting: {type(val)}")
            fig_axs.append((f, a))
        return fig_axs
