"""L6 — plotting helpers (matplotlib optional).

Parity: torchmetrics ``utilities/plot.py`` (plot_single_or_multi_val,
plot_confusion_matrix, plot_curve).
"""
from __future__ import annotations

from itertools import product
from typing import Any, List, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor

from metrics_amd.utilities.imports import _MATPLOTLIB_AVAILABLE

if _MATPLOTLIB_AVAILABLE:
    import matplotlib
    import matplotlib.axes
    import matplotlib.pyplot as plt

    _AX_TYPE = "matplotlib.axes.Axes"
    _PLOT_OUT_TYPE = Tuple["plt.Figure", Union["matplotlib.axes.Axes", Any]]
else:
    _AX_TYPE = object
    _PLOT_OUT_TYPE = tuple  # type: ignore[misc]


def _error_on_missing_matplotlib() -> None:
    if not _MATPLOTLIB_AVAILABLE:
        raise ModuleNotFoundError(
            "Plot function expects `matplotlib` to be installed, which is not available in this environment."
        )


def plot_single_or_multi_val(
    val,
    ax=None,
    higher_is_better: Optional[bool] = None,
    name: Optional[str] = None,
    lower_bound: Optional[float] = None,
    upper_bound: Optional[float] = None,
    legend_name: Optional[str] = None,
):
    """Plot a single scalar result, a per-class vector, or a sequence of results over steps."""
    _error_on_missing_matplotlib()
    fig, ax = plt.subplots() if ax is None else (None, ax)
    ax.get_xaxis().set_visible(False)

    if isinstance(val, Tensor):
        if val.numel() == 1:
            ax.plot([val.detach().cpu()], marker="o", markersize=10)
        else:
            var = val.detach().cpu()
            ax.plot(var, marker="o", markersize=10)
            ax.get_xaxis().set_visible(True)
    elif isinstance(val, dict):
        for i, (k, v) in enumerate(val.items()):
            v = v.detach().cpu()
            if v.numel() == 1:
                ax.plot([v], marker="o", markersize=10, label=k)
            else:
                ax.plot(v, marker="o", markersize=10, label=k)
                ax.get_xaxis().set_visible(True)
        ax.legend()
    elif isinstance(val, Sequence):
        n_steps = len(val)
        if isinstance(val[0], dict):
            val_flat = {k: torch.stack([val[i][k].detach().cpu() for i in range(n_steps)]) for k in val[0]}
            for k, v in val_flat.items():
                ax.plot(range(n_steps), v, marker="o", markersize=10, label=k)
            ax.legend()
        else:
            vals = torch.stack([torch.as_tensor(v).detach().cpu() for v in val], 0)
            ax.plot(range(n_steps), vals, marker="o", markersize=10)
        ax.get_xaxis().set_visible(True)
        ax.set_xlabel("Step")
    else:
        raise ValueError(f"Unknown format of input value: {type(val)}")

    if lower_bound is not None or upper_bound is not None:
        ylim = ax.get_ylim()
        ax.set_ylim(
            bottom=lower_bound if lower_bound is not None else ylim[0],
            top=upper_bound if upper_bound is not None else ylim[1],
        )
    if name is not None:
        ax.set_ylabel(name)
    return fig, ax


def trim_axs(axs, nb: int):
    """Hide extra axes when the grid is larger than the number of plots."""
    if isinstance(axs, Sequence) or hasattr(axs, "flat"):
        axs = axs.flat if hasattr(axs, "flat") else axs
        axs = list(axs)
        for ax in axs[nb:]:
            ax.set_visible(False)
        return axs[:nb]
    return axs


def plot_confusion_matrix(
    confmat: Tensor,
    ax=None,
    add_text: bool = True,
    labels: Optional[List[Union[int, str]]] = None,
    cmap=None,
):
    """Heatmap plot of a (C,C) or (N,2,2) confusion matrix."""
    _error_on_missing_matplotlib()
    multilabel = confmat.ndim == 3
    if multilabel:
        nb, n_classes = confmat.shape[0], 2
        rows, cols = _get_col_row_split(nb)
    else:
        nb, n_classes, rows, cols = 1, confmat.shape[0], 1, 1
        confmat = confmat[None]

    if labels is not None and not multilabel and len(labels) != n_classes:
        raise ValueError(
            "Expected number of elements in arg `labels` to match number of labels in confmat but "
            f"got {len(labels)} and {n_classes}"
        )

    if multilabel:
        fig_label = labels or np_arange(nb)
        labels = [0, 1]
    else:
        fig_label = None
        labels = labels or np_arange(n_classes)

    if ax is None:
        fig, axs = plt.subplots(nrows=rows, ncols=cols)
    else:
        fig, axs = None, ax

    axs = trim_axs(axs, nb) if nb > 1 else [axs]
    for i in range(nb):
        ax_ = axs[i] if nb > 1 else axs[0]
        if fig_label is not None:
            ax_.set_title(f"Label {fig_label[i]}", fontsize=15)
        ax_.imshow(confmat[i].cpu().detach(), cmap=cmap)
        ax_.set_xlabel("Predicted class", fontsize=15)
        ax_.set_ylabel("True class", fontsize=15)
        ax_.set_xticks(list(range(n_classes)))
        ax_.set_yticks(list(range(n_classes)))
        ax_.set_xticklabels(labels, rotation=45, fontsize=10)
        ax_.set_yticklabels(labels, rotation=25, fontsize=10)

        if add_text:
            for ii, jj in product(range(n_classes), range(n_classes)):
                val = confmat[i, ii, jj]
                ax_.text(jj, ii, str(round(val.item(), 2)), ha="center", va="center", fontsize=15)

    return fig, axs


def np_arange(n: int) -> list:
    return list(range(n))


def _get_col_row_split(n: int) -> Tuple[int, int]:
    """Split n plots into a (rows, cols) grid close to square."""
    nsq = int(n**0.5)
    if nsq * nsq == n:
        return nsq, nsq
    if n <= nsq * (nsq + 1):
        return nsq, nsq + 1
    return nsq + 1, nsq + 1


def plot_curve(
    curve: Tuple[Tensor, Tensor, Tensor],
    score: Optional[Tensor] = None,
    ax=None,
    label_names: Optional[Tuple[str, str]] = None,
    legend_name: Optional[str] = None,
    name: Optional[str] = None,
):
    """Plot a (x, y, thresholds)-style curve (ROC / PR)."""
    _error_on_missing_matplotlib()
    if len(curve) < 2:
        raise ValueError("Expected 2 or more elements in curve but got {len(curve)}")
    x, y = curve[:2]

    fig, ax = plt.subplots() if ax is None else (None, ax)

    if isinstance(x, Tensor) and isinstance(y, Tensor) and x.ndim == 1 and y.ndim == 1:
        label = f"AUC={score.item():0.3f}" if score is not None else None
        ax.plot(x.detach().cpu(), y.detach().cpu(), linestyle="-", linewidth=2, label=label)
        if label_names is not None:
            ax.set_xlabel(label_names[0])
            ax.set_ylabel(label_names[1])
        if label is not None:
            ax.legend()
    elif (isinstance(x, list) and isinstance(y, list)) or (
        isinstance(x, Tensor) and isinstance(y, Tensor) and x.ndim == 2 and y.ndim == 2
    ):
        for i, (x_, y_) in enumerate(zip(x, y)):
            label = f"{legend_name}_{i}" if legend_name is not None else str(i)
            if score is not None:
                label += f" AUC={score[i].item():0.3f}"
            ax.plot(x_.detach().cpu(), y_.detach().cpu(), label=label)
        ax.legend()
    else:
        raise ValueError(
            f"Unknown format of input arguments: {type(x)} and {type(y)}. Expected either Tensors or list of Tensors."
        )

    ax.grid(True)
    if name is not None:
        ax.set_title(name)
    return fig, ax
