"""Cohen's kappa. Parity: torchmetrics ``functional/classification/cohen_kappa.py``."""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from metrics_amd.utilities.enums import ClassificationTaskNoMultilabel
from metrics_amd.functional.classification.confusion_matrix import (
    _binary_confusion_matrix_arg_validation,
    _binary_confusion_matrix_format,
    _binary_confusion_matrix_tensor_validation,
    _binary_confusion_matrix_update,
    _multiclass_confusion_matrix_arg_validation,
    _multiclass_confusion_matrix_format,
    _multiclass_confusion_matrix_tensor_validation,
    _multiclass_confusion_matrix_update,
)


def _cohen_kappa_reduce(confmat: Tensor, weights: Optional[str] = None) -> Tensor:
    """Kappa from a confusion matrix with optional linear/quadratic disagreement weighting."""
    if weights is None and confmat.is_cuda and confmat.ndim == 2 and confmat.dtype == torch.long:
        from metrics_amd.ops import _hip

        if _hip.hip_available():
            return _hip.confmat_scalars(confmat)[1].clone()
    confmat = confmat.float()
    num_classes = confmat.shape[0]
    sum0 = confmat.sum(dim=0, keepdim=True)
    sum1 = confmat.sum(dim=1, keepdim=True)

    if weights in (None, "quadratic") and confmat.is_cuda:
        # separable identities avoid materializing the (C,C) weight and
        # expected matrices (the (C,C) outer product dominated compute time):
        #   None:      sum(w*cm) = s - trace;  sum(w*E) = s - sum(tk*pk)/s
        #   quadratic: w_ij=(i-j)^2 = i^2 - 2ij + j^2 (separable in i, j)
        tk = sum1.flatten()
        pk = sum0.flatten()
        s_total = pk.sum()
        if weights is None:
            obs = s_total - torch.diagonal(confmat).sum()
            exp = s_total - (tk * pk).sum() / s_total
        else:
            idx = torch.arange(num_classes, dtype=confmat.dtype, device=confmat.device)
            i2cm_rows = (confmat * (idx**2).unsqueeze(1)).sum()  # sum i^2 cm_ij
            j2cm_cols = (confmat * (idx**2).unsqueeze(0)).sum()
            ijcm = (idx.unsqueeze(1) * idx.unsqueeze(0) * confmat).sum()
            obs = i2cm_rows + j2cm_cols - 2 * ijcm
            # sum(w*E) with E = outer(tk,pk)/s and w_ij = i^2 - 2ij + j^2:
            exp = (idx**2 * tk).sum() + (idx**2 * pk).sum() - 2 * (idx * tk).sum() * (idx * pk).sum() / s_total
        return 1 - obs / exp

    expected = sum1 @ sum0 / sum0.sum()  # outer product

    if weights is None:
        w_mat = torch.ones_like(confmat).flatten()
        w_mat[:: num_classes + 1] = 0
        w_mat = w_mat.reshape(num_classes, num_classes)
    elif weights in ("linear", "quadratic"):
        w_mat = torch.zeros_like(confmat)
        w_mat += torch.arange(num_classes, dtype=w_mat.dtype, device=w_mat.device)
        if weights == "linear":
            w_mat = torch.abs(w_mat - w_mat.T)
        else:
            w_mat = torch.pow(w_mat - w_mat.T, 2.0)
    else:
        raise ValueError(
            f"Received {weights} for argument ``weights`` but should be either None, 'linear' or 'quadratic'"
        )
    k = torch.sum(w_mat * confmat) / torch.sum(w_mat * expected)
    return 1 - k


def _cohen_kappa_arg_validation(weights: Optional[str]) -> None:
    allowed_weights = ("linear", "quadratic", "none", None)
    if weights not in allowed_weights:
        raise ValueError(f"Expected argument `weight` to be one of {allowed_weights}, but got {weights}.")


def binary_cohen_kappa(
    preds: Tensor,
    target: Tensor,
    threshold: float = 0.5,
    weights: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Cohen's kappa for binary tasks."""
    if validate_args:
        _binary_confusion_matrix_arg_validation(threshold, ignore_index, normalize=None)
        _cohen_kappa_arg_validation(weights)
        _binary_confusion_matrix_tensor_validation(preds, target, ignore_index)
    preds, target = _binary_confusion_matrix_format(preds, target, threshold, ignore_index)
    confmat = _binary_confusion_matrix_update(preds, target)
    return _cohen_kappa_reduce(confmat, weights)


def multiclass_cohen_kappa(
    preds: Tensor,
    target: Tensor,
    num_classes: int,
    weights: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Cohen's kappa for multiclass tasks."""
    if validate_args:
        _multiclass_confusion_matrix_arg_validation(num_classes, ignore_index, normalize=None)
        _cohen_kappa_arg_validation(weights)
        _multiclass_confusion_matrix_tensor_validation(preds, target, num_classes, ignore_index)
    preds, target = _multiclass_confusion_matrix_format(preds, target, ignore_index)
    confmat = _multiclass_confusion_matrix_update(preds, target, num_classes)
    return _cohen_kappa_reduce(confmat, weights)


def cohen_kappa(
    preds: Tensor,
    target: Tensor,
    task: str,
    threshold: float = 0.5,
    num_classes: Optional[int] = None,
    weights: Optional[str] = None,
    ignore_index: Optional[int] = None,
    validate_args: bool = True,
) -> Tensor:
    """Task-dispatching Cohen's kappa."""
    task = ClassificationTaskNoMultilabel.from_str(task)
    if task == ClassificationTaskNoMultilabel.BINARY:
        return binary_cohen_kappa(preds, target, threshold, weights, ignore_index, validate_args)
    if task == ClassificationTaskNoMultilabel.MULTICLASS:
        if not isinstance(num_classes, int):
            raise ValueError(f"`num_classes` is expected to be `int` but `{type(num_classes)} was passed.`")
        return multiclass_cohen_kappa(preds, target, num_classes, weights, ignore_index, validate_args)
    raise ValueError(f"Not handled value: {task}")
