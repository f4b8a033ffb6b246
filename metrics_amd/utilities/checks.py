"""Input validation helpers.

Parity: torchmetrics ``utilities/checks.py`` (_check_same_shape,
check_forward_full_state_property). The legacy multi-format input classifier
is not carried over: the new classification stack is task-explicit
(binary/multiclass/multilabel classes), matching the reference's modern API.
"""
from __future__ import annotations

from time import perf_counter
from typing import Any, Callable, Dict, Optional

import torch
from torch import Tensor

from metrics_amd.utilities.prints import rank_zero_info


def _check_same_shape(preds: Tensor, target: Tensor) -> None:
    """Raise if ``preds`` and ``target`` have different shapes."""
    if preds.shape != target.shape:
        raise RuntimeError(
            f"Predictions and targets are expected to have the same shape, but got {preds.shape} and {target.shape}."
        )


def _allclose_recursive(res1: Any, res2: Any, atol: float = 1e-8) -> bool:
    """Recursively asserting that two results are within a certain tolerance."""
    if isinstance(res1, Tensor):
        return torch.allclose(res1, res2, atol=atol)
    if isinstance(res1, str):
        return res1 == res2
    if isinstance(res1, (list, tuple)):
        return all(_allclose_recursive(r1, r2) for r1, r2 in zip(res1, res2))
    if isinstance(res1, dict):
        return all(_allclose_recursive(res1[k], res2[k]) for k in res1)
    return res1 == res2


def check_forward_full_state_property(
    metric_class: type,
    init_args: Optional[Dict[str, Any]] = None,
    input_args: Optional[Dict[str, Any]] = None,
    num_update_to_compare: tuple = (10, 100, 1000),
    reps: int = 5,
) -> None:
    """Benchmark whether ``full_state_update=False`` is safe and faster for a metric.

    Instantiates the metric with ``full_state_update`` True and False, checks
    both give identical forward results, and reports timing.
    """
    init_args = init_args or {}
    input_args = input_args or {}

    class FullState(metric_class):  # type: ignore[misc, valid-type]
        full_state_update = True

    class PartState(metric_class):  # type: ignore[misc, valid-type]
        full_state_update = False

    fullstate = FullState(**init_args)
    partstate = PartState(**init_args)

    equal = True
    try:
        for _ in range(num_update_to_compare[0]):
            equal = equal & _allclose_recursive(fullstate(**input_args), partstate(**input_args))
        res1 = fullstate.compute()
        res2 = partstate.compute()
        equal = equal & _allclose_recursive(res1, res2)
    except Exception:
        equal = False

    mean_time_full, mean_time_part = [], []
    for n in num_update_to_compare:
        for mlist, m in ((mean_time_full, FullState(**init_args)), (mean_time_part, PartState(**init_args))):
            times = []
            for _ in range(reps):
                start = perf_counter()
                for _ in range(n):
                    m(**input_args)
                times.append(perf_counter() - start)
                m.reset()
            mlist.append(sum(times) / len(times))

    rank_zero_info(f"Full state for {num_update_to_compare} steps took: {mean_time_full}")
    rank_zero_info(f"Partial state for {num_update_to_compare} steps took: {mean_time_part}")

    faster = all(p <= f for p, f in zip(mean_time_part, mean_time_full))

    if not equal:
        raise ValueError(
            "The metric does not provide the same result when using `full_state_update=False` — it cannot be disabled."
        )
    if equal and faster:
        rank_zero_info("The metric can safely set `full_state_update=False` (equal results, faster).")


def _check_retrieval_functional_inputs(
    preds: Tensor,
    target: Tensor,
    allow_non_binary_target: bool = False,
) -> tuple:
    """Validate (preds, target) for functional retrieval metrics; returns float preds + processed target."""
    if preds.shape != target.shape:
        raise ValueError("`preds` and `target` must be of the same shape")
    if not preds.numel() or not preds.size():
        raise ValueError("`preds` and `target` must be non-empty and non-scalar tensors")
    return _check_retrieval_target_and_prediction_types(preds, target, allow_non_binary_target)


def _check_retrieval_inputs(
    indexes: Tensor,
    preds: Tensor,
    target: Tensor,
    allow_non_binary_target: bool = False,
    ignore_index: Optional[int] = None,
) -> tuple:
    """Validate (indexes, preds, target) for retrieval Metric modules."""
    if indexes.shape != preds.shape or preds.shape != target.shape:
        raise ValueError("`indexes`, `preds` and `target` must be of the same shape")
    if indexes.dtype is not torch.long:
        raise ValueError("`indexes` must be a tensor of long integers")
    if not indexes.numel() or not indexes.size():
        raise ValueError("`indexes`, `preds` and `target` must be non-empty and non-scalar tensors")

    if ignore_index is not None:
        valid_positions = target != ignore_index
        indexes, preds, target = indexes[valid_positions], preds[valid_positions], target[valid_positions]

    preds, target = _check_retrieval_target_and_prediction_types(preds, target, allow_non_binary_target)
    return indexes.long().flatten(), preds, target


def _check_retrieval_target_and_prediction_types(
    preds: Tensor, target: Tensor, allow_non_binary_target: bool
) -> tuple:
    if target.dtype not in (torch.bool, torch.long, torch.int) and not torch.is_floating_point(target):
        raise ValueError("`target` must be a tensor of booleans, integers or floats")
    if not preds.is_floating_point():
        raise ValueError("`preds` must be a tensor of floats")
    if not allow_non_binary_target and (target.max() > 1 or target.min() < 0):
        raise ValueError("`target` must contain `binary` values")
    target = target.float() if target.is_floating_point() else target.long()
    return preds.float().flatten(), target.flatten()
