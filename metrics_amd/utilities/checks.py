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
    try:  # a failure here usually means the partial path needs the full state
        for _ in range(num_update_to_compare[0]):
            equal = equal and bool(_allclose_recursive(fullstate(**input_args), partstate(**input_args)))
    except RuntimeError:
        equal = False
    res1 = fullstate.compute()
    try:
        res2 = partstate.compute()
    except RuntimeError:
        equal = False
        res2 = None
    if equal:
        equal = bool(_allclose_recursive(res1, res2))

    if not equal:  # results diverge — the metric needs the full state in forward
        print("Recommended setting `full_state_update=True`")
        return

    res = torch.zeros(2, len(num_update_to_compare), reps)
    for i, metric in enumerate([fullstate, partstate]):
        for j, n in enumerate(num_update_to_compare):
            for r in range(reps):
                start = perf_counter()
                for _ in range(n):
                    metric(**input_args)
                res[i, j, r] = perf_counter() - start
                metric.reset()

    mean = torch.mean(res, -1)
    std = torch.std(res, -1)
    for j, n in enumerate(num_update_to_compare):
        print(f"Full state for {n} steps took: {mean[0, j]}+-{std[0, j]:0.3f}")
        print(f"Partial state for {n} steps took: {mean[1, j]:0.3f}+-{std[1, j]:0.3f}")

    faster = bool((mean[1, -1] < mean[0, -1]).item())
    print(f"Recommended setting `full_state_update={not faster}`")


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
