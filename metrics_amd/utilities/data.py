"""L0 tensor primitives.

Parity: torchmetrics ``utilities/data.py`` (dim_zero_* reductions, to_onehot,
select_topk, _bincount, _flexible_bincount, _cumsum, interp).

MI355X-first deltas vs the reference:
- ``_bincount`` dispatches to the in-tree HIP LDS-privatized histogram kernel on
  GPU tensors (integer atomics => deterministic by construction), so the
  reference's deterministic/XLA/MPS fallback ladder is gone.
- no branches on torch version / backend.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Union

import torch
from torch import Tensor

METRIC_EPS = 1e-6


def dim_zero_cat(x: Union[Tensor, List[Tensor]]) -> Tensor:
    """Concatenate a (list of) tensor(s) along dim 0."""
    if isinstance(x, Tensor):
        return x
    x = [y.unsqueeze(0) if y.numel() == 1 and y.ndim == 0 else y for y in x]
    if not x:  # empty list
        raise ValueError("No samples to concatenate")
    return torch.cat(x, dim=0)


def dim_zero_sum(x: Tensor) -> Tensor:
    return torch.sum(x, dim=0)


def dim_zero_mean(x: Tensor) -> Tensor:
    return torch.mean(x.float(), dim=0)


def dim_zero_max(x: Tensor) -> Tensor:
    return torch.max(x, dim=0).values


def dim_zero_min(x: Tensor) -> Tensor:
    return torch.min(x, dim=0).values


def _flatten(x: Sequence) -> list:
    """Flatten one level of nesting."""
    return [item for sublist in x for item in sublist]


def _flatten_dict(x: dict) -> tuple:
    """Flatten one level of nested dicts; returns (flat_dict, any_duplicates)."""
    new_dict = {}
    duplicates = False
    for key, value in x.items():
        if isinstance(value, dict):
            for k, v in value.items():
                if k in new_dict:
                    duplicates = True
                new_dict[k] = v
        else:
            if key in new_dict:
                duplicates = True
            new_dict[key] = value
    return new_dict, duplicates


def to_onehot(label_tensor: Tensor, num_classes: Optional[int] = None) -> Tensor:
    """Convert a dense label tensor ``(N, ...)`` to one-hot ``(N, C, ...)``.

    ``num_classes`` defaults to ``max(labels) + 1`` (reference utilities/data.py:81).
    """
    if label_tensor.ndim == 0:
        label_tensor = label_tensor.unsqueeze(0)
    if num_classes is None:
        num_classes = int(label_tensor.max().detach().item() + 1)
    shape = label_tensor.shape
    out = torch.zeros(
        shape[0], num_classes, *shape[1:], dtype=label_tensor.dtype, device=label_tensor.device
    )
    index = label_tensor.long().unsqueeze(1).expand_as(out.narrow(1, 0, 1)).clamp_(0, num_classes - 1)
    return out.scatter_(1, index, 1.0)


def to_categorical(x: Tensor, argmax_dim: int = 1) -> Tensor:
    """Convert a probability tensor to dense labels via argmax (reference utilities/data.py:151)."""
    return torch.argmax(x, dim=argmax_dim)


def select_topk(prob_tensor: Tensor, topk: int = 1, dim: int = 1) -> Tensor:
    """Binary one-hot mask of the ``topk`` highest entries along ``dim``.

    Top-1 uses argmax (cheaper than topk); output dtype is int.
    """
    if topk == 1:
        topk_tensor = torch.zeros_like(prob_tensor, dtype=torch.int)
        idx = prob_tensor.argmax(dim=dim, keepdim=True)
        return topk_tensor.scatter_(dim, idx, 1)
    zeros = torch.zeros_like(prob_tensor, dtype=torch.int)
    return zeros.scatter_(dim, prob_tensor.topk(k=topk, dim=dim).indices, 1)


def _bincount(x: Tensor, minlength: Optional[int] = None) -> Tensor:
    """Histogram of a non-negative integer tensor.

    GPU: in-tree HIP kernel (LDS-privatized bins, integer atomics =>
    deterministic). CPU: torch.bincount.
    """
    if minlength is None:
        minlength = len(torch.unique(x))
    if x.is_cuda:
        from metrics_amd.ops import hip_bincount

        return hip_bincount(x, minlength)
    return torch.bincount(x.flatten(), minlength=minlength)


def _flexible_bincount(x: Tensor) -> Tensor:
    """Bincount over arbitrary (possibly negative / sparse) integer values."""
    # map values to a dense [0, n_unique) index space first
    _, inverse, counts = torch.unique(x, return_inverse=True, return_counts=True)
    del inverse
    return counts


def _cumsum(x: Tensor, dim: int = 0, dtype: Optional[torch.dtype] = None) -> Tensor:
    return torch.cumsum(x, dim=dim, dtype=dtype)


def interp(x: Tensor, xp: Tensor, fp: Tensor) -> Tensor:
    """1-D linear interpolation (numpy.interp semantics for increasing xp)."""
    slopes = (fp[1:] - fp[:-1]) / (xp[1:] - xp[:-1])
    locs = torch.searchsorted(xp, x)
    locs = locs.clamp(1, len(xp) - 1) - 1
    return slopes[locs] * (x - xp[locs]) + fp[locs]


def apply_to_collection(data, dtype, function, *args, **kwargs):
    """Recursively apply ``function`` to all elements of type ``dtype`` in a collection."""
    if isinstance(data, dtype):
        return function(data, *args, **kwargs)
    if isinstance(data, dict):
        return {k: apply_to_collection(v, dtype, function, *args, **kwargs) for k, v in data.items()}
    if isinstance(data, tuple) and hasattr(data, "_fields"):  # namedtuple
        return type(data)(*(apply_to_collection(d, dtype, function, *args, **kwargs) for d in data))
    if isinstance(data, (list, tuple)):
        return type(data)(apply_to_collection(d, dtype, function, *args, **kwargs) for d in data)
    return data


def allclose(tensor1: Tensor, tensor2: Tensor) -> bool:
    """allclose that tolerates dtype mismatch."""
    if tensor1.dtype != tensor2.dtype:
        tensor2 = tensor2.to(dtype=tensor1.dtype)
    return torch.allclose(tensor1, tensor2)
