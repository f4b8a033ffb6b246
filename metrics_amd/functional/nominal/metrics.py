"""Nominal association metrics (functional).

Parity: torchmetrics ``functional/nominal/*`` — Cramer's V, Pearson's
contingency coefficient, Tschuprow's T, Theil's U, Fleiss kappa.
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from metrics_amd.utilities.data import _bincount


def _nominal_input_validation(nan_strategy: str, nan_replace_value: Optional[float]) -> None:
    if nan_strategy not in ("replace", "drop"):
        raise ValueError(
            f"Argument `nan_strategy` is expected to be one of `['replace', 'drop']`, but got {nan_strategy}"
        )
    if nan_strategy == "replace" and not isinstance(nan_replace_value, (int, float)):
        raise ValueError(
            "Argument `nan_replace_value` is expected to be of a type `int` or `float` when `nan_strategy = 'replace`, "
            f"but got {nan_replace_value}"
        )


def _handle_nan_in_data(preds: Tensor, target: Tensor, nan_strategy: str = "replace", nan_replace_value=0.0):
    if nan_strategy == "replace":
        return preds.nan_to_num(nan_replace_value), target.nan_to_num(nan_replace_value)
    rows_contain_nan = torch.logical_or(preds.isnan(), target.isnan())
    return preds[~rows_contain_nan], target[~rows_contain_nan]


def _compute_contingency(preds: Tensor, target: Tensor) -> Tensor:
    """Contingency matrix over dense-mapped label values."""
    _, p_idx = torch.unique(preds, return_inverse=True)
    _, t_idx = torch.unique(target, return_inverse=True)
    num_p = int(p_idx.max()) + 1
    num_t = int(t_idx.max()) + 1
    flat = t_idx * num_p + p_idx
    return _bincount(flat, minlength=num_p * num_t).reshape(num_t, num_p).float()


def _drop_empty_rows_and_cols(confmat: Tensor) -> Tensor:
    confmat = confmat[confmat.sum(1) != 0]
    return confmat[:, confmat.sum(0) != 0]


def _format_inputs(preds: Tensor, target: Tensor, nan_strategy: str, nan_replace_value) -> Tensor:
    preds = preds.argmax(1) if preds.ndim == 2 else preds
    target = target.argmax(1) if target.ndim == 2 else target
    preds, target = _handle_nan_in_data(preds, target, nan_strategy, nan_replace_value)
    return _drop_empty_rows_and_cols(_compute_contingency(preds, target))


def cramers_v(
    preds: Tensor,
    target: Tensor,
    bias_correction: bool = True,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Cramer's V statistic measuring association between two categorical variables."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    confmat = _format_inputs(preds, target, nan_strategy, nan_replace_value)
    cm_sum = confmat.sum()
    chi_squared = _chi_squared(confmat)
    phi_squared = chi_squared / cm_sum
    num_rows, num_cols = confmat.shape

    if bias_correction:
        phi_squared_corrected = torch.max(
            torch.tensor(0.0, device=confmat.device), phi_squared - ((num_rows - 1) * (num_cols - 1)) / (cm_sum - 1)
        )
        rows_corrected = num_rows - (num_rows - 1) ** 2 / (cm_sum - 1)
        cols_corrected = num_cols - (num_cols - 1) ** 2 / (cm_sum - 1)
        if min(rows_corrected, cols_corrected) == 1:
            import warnings

            warnings.warn(
                "Unable to compute Cramer's V using bias correction. Please consider to set `bias_correction=False`.",
                UserWarning, stacklevel=2,
            )
            return torch.tensor(float("nan"), device=confmat.device)
        cramers_v_value = torch.sqrt(phi_squared_corrected / min(rows_corrected - 1, cols_corrected - 1))
    else:
        cramers_v_value = torch.sqrt(phi_squared / min(num_rows - 1, num_cols - 1))
    return cramers_v_value.clamp(0.0, 1.0)


def _chi_squared(confmat: Tensor) -> Tensor:
    cm_sum = confmat.sum()
    expected = confmat.sum(1, keepdim=True) @ confmat.sum(0, keepdim=True) / cm_sum
    return ((confmat - expected) ** 2 / expected).sum()


def pearsons_contingency_coefficient(
    preds: Tensor,
    target: Tensor,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Pearson's contingency coefficient."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    confmat = _format_inputs(preds, target, nan_strategy, nan_replace_value)
    cm_sum = confmat.sum()
    chi_squared = _chi_squared(confmat)
    phi_squared = chi_squared / cm_sum
    return torch.sqrt(phi_squared / (1 + phi_squared)).clamp(0.0, 1.0)


def tschuprows_t(
    preds: Tensor,
    target: Tensor,
    bias_correction: bool = True,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Tschuprow's T statistic."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    confmat = _format_inputs(preds, target, nan_strategy, nan_replace_value)
    cm_sum = confmat.sum()
    chi_squared = _chi_squared(confmat)
    phi_squared = chi_squared / cm_sum
    num_rows, num_cols = confmat.shape

    if bias_correction:
        phi_squared_corrected = torch.max(
            torch.tensor(0.0, device=confmat.device), phi_squared - ((num_rows - 1) * (num_cols - 1)) / (cm_sum - 1)
        )
        rows_corrected = num_rows - (num_rows - 1) ** 2 / (cm_sum - 1)
        cols_corrected = num_cols - (num_cols - 1) ** 2 / (cm_sum - 1)
        if min(rows_corrected, cols_corrected) == 1:
            import warnings

            warnings.warn(
                "Unable to compute Tschuprow's T using bias correction. Please consider to set `bias_correction=False`.",
                UserWarning, stacklevel=2,
            )
            return torch.tensor(float("nan"), device=confmat.device)
        t = torch.sqrt(phi_squared_corrected / torch.sqrt((rows_corrected - 1) * (cols_corrected - 1)))
    else:
        n_rows_t = torch.tensor(num_rows - 1, dtype=torch.float32, device=confmat.device)
        n_cols_t = torch.tensor(num_cols - 1, dtype=torch.float32, device=confmat.device)
        t = torch.sqrt(phi_squared / torch.sqrt(n_rows_t * n_cols_t))
    return t.clamp(0.0, 1.0)


def theils_u(
    preds: Tensor,
    target: Tensor,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Theil's U (uncertainty coefficient) U(preds|target)."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    confmat = _format_inputs(preds, target, nan_strategy, nan_replace_value)
    total = confmat.sum()

    # H(X) where X = preds (columns)
    p_x = confmat.sum(0) / total
    p_x = p_x[p_x > 0]
    h_x = -(p_x * torch.log(p_x)).sum()

    # H(X|Y) over targets (rows)
    p_y = confmat.sum(1) / total
    h_xy = torch.tensor(0.0, device=confmat.device)
    for r in range(confmat.shape[0]):
        row = confmat[r]
        row_total = row.sum()
        if row_total == 0:
            continue
        p = row[row > 0] / row_total
        h_xy = h_xy - p_y[r] * (p * torch.log(p)).sum()

    if h_x == 0:
        return torch.tensor(1.0, device=confmat.device)
    return ((h_x - h_xy) / h_x).clamp(0.0, 1.0)


def fleiss_kappa(ratings: Tensor, mode: str = "counts") -> Tensor:
    """Fleiss' kappa for inter-rater agreement.

    ``ratings``: (n_subjects, n_categories) count matrix when mode='counts',
    or (n_subjects, n_categories, n_raters) probabilities when mode='probs'.
    """
    if mode == "probs":
        if ratings.ndim != 3 or not ratings.is_floating_point():
            raise ValueError("If argument `mode` is 'probs', ratings must have 3 dimensions with the format [n_samples, n_categories, n_raters] and be floating point")
        # verbatim reference transform (functional/nominal/fleiss_kappa.py:27-35)
        ratings = ratings.argmax(dim=1)
        one_hot = torch.nn.functional.one_hot(ratings, num_classes=ratings.shape[1]).permute(0, 2, 1)
        ratings = one_hot.sum(dim=-1)
    elif mode == "counts":
        if ratings.ndim != 2 or ratings.is_floating_point():
            raise ValueError("If argument `mode` is `counts`, ratings must have 2 dimensions with the format [n_samples, n_categories] and be none floating point")
    else:
        raise ValueError("Argument `mode` should be one of 'counts' or 'probs'")

    counts = ratings.float()
    total = counts.shape[0]
    # reference semantics: unequal rater counts allowed, normalized by the MAX
    num_raters = counts.sum(1).max()
    p_i = counts.sum(dim=0) / (total * num_raters)
    p_j = ((counts**2).sum(dim=1) - num_raters) / (num_raters * (num_raters - 1))
    p_bar = p_j.mean()
    pe_bar = (p_i**2).sum()
    return (p_bar - pe_bar) / (1 - pe_bar + 1e-5)


def _matrix_over_columns(matrix: Tensor, pair_fn, symmetric: bool) -> Tensor:
    """Pairwise nominal statistic over the columns of a (N, V) data matrix."""
    if matrix.ndim != 2:
        raise ValueError(f"Expected `matrix` to be a 2D tensor of shape (observations, variables), got {matrix.shape}")
    num_variables = matrix.shape[1]
    out = torch.ones(num_variables, num_variables, device=matrix.device)
    for i in range(num_variables):
        for j in range(i + 1, num_variables):
            x, y = matrix[:, i], matrix[:, j]
            out[i, j] = pair_fn(x, y)
            out[j, i] = out[i, j] if symmetric else pair_fn(y, x)
    return out


def cramers_v_matrix(
    matrix: Tensor,
    bias_correction: bool = True,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Pairwise Cramer's V over the columns of a categorical data matrix.

    Parity: reference functional/nominal/cramers.py:141.
    """
    _nominal_input_validation(nan_strategy, nan_replace_value)
    return _matrix_over_columns(
        matrix, lambda x, y: cramers_v(x, y, bias_correction, nan_strategy, nan_replace_value), symmetric=True
    )


def pearsons_contingency_coefficient_matrix(
    matrix: Tensor,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Pairwise Pearson's contingency coefficient over data-matrix columns (reference functional/nominal/pearson.py)."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    return _matrix_over_columns(
        matrix, lambda x, y: pearsons_contingency_coefficient(x, y, nan_strategy, nan_replace_value), symmetric=True
    )


def tschuprows_t_matrix(
    matrix: Tensor,
    bias_correction: bool = True,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Pairwise Tschuprow's T over data-matrix columns (reference functional/nominal/tschuprows.py)."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    return _matrix_over_columns(
        matrix, lambda x, y: tschuprows_t(x, y, bias_correction, nan_strategy, nan_replace_value), symmetric=True
    )


def theils_u_matrix(
    matrix: Tensor,
    nan_strategy: str = "replace",
    nan_replace_value: Optional[float] = 0.0,
) -> Tensor:
    """Pairwise (asymmetric) Theil's U over data-matrix columns (reference functional/nominal/theils_u.py)."""
    _nominal_input_validation(nan_strategy, nan_replace_value)
    return _matrix_over_columns(
        matrix, lambda x, y: theils_u(x, y, nan_strategy, nan_replace_value), symmetric=False
    )
