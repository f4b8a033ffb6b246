"""Functional BERTScore and InfoLM with user-supplied local transformer models.

Parity: reference ``functional/text/bert.py:260`` (``bert_score``) and
``functional/text/infolm.py:546`` (``infolm``). The reference downloads
pretrained weights by name; offline, a local ``model``/``user_tokenizer``
(HuggingFace-style) must be supplied — without one these raise
``ModuleNotFoundError`` just like the reference does without its deps.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple, Union

import torch
from torch import Tensor

_ALLOWED_INFORMATION_MEASURE = (
    "kl_divergence",
    "alpha_divergence",
    "beta_divergence",
    "ab_divergence",
    "renyi_divergence",
    "l1_distance",
    "l2_distance",
    "l_infinity_distance",
    "fisher_rao_distance",
)


@torch.no_grad()
def _embed(model, tokenizer, texts: List[str], max_length: int, device) -> Tuple[Tensor, Tensor]:
    enc = tokenizer(texts, return_tensors="pt", padding=True, truncation=True, max_length=max_length)
    if device is not None:
        enc = {k: v.to(device) for k, v in enc.items()}
    out = model(**enc)
    emb = out.last_hidden_state if hasattr(out, "last_hidden_state") else out[0]
    emb = emb / emb.norm(dim=-1, keepdim=True).clamp(min=1e-12)
    return emb, enc["attention_mask"].bool()


def bert_score(
    preds: Union[str, Sequence[str], Dict[str, Tensor]],
    target: Union[str, Sequence[str], Dict[str, Tensor]],
    model_name_or_path: Optional[str] = None,
    num_layers: Optional[int] = None,
    all_layers: bool = False,
    model: Optional[torch.nn.Module] = None,
    user_tokenizer: Any = None,
    user_forward_fn: Optional[Callable] = None,
    verbose: bool = False,
    idf: bool = False,
    device: Optional[Union[str, torch.device]] = None,
    max_length: int = 512,
    batch_size: int = 64,
    num_threads: int = 0,
    return_hash: bool = False,
    lang: str = "en",
    rescale_with_baseline: bool = False,
    baseline_path: Optional[str] = None,
    baseline_url: Optional[str] = None,
    truncation: bool = False,
) -> Dict[str, Union[Tensor, List[float], str]]:
    """BERTScore: greedy cosine matching of contextual embeddings (P/R/F1)."""
    if model is None or user_tokenizer is None:
        raise ModuleNotFoundError(
            "`bert_score` needs a local transformer model + tokenizer: pass `model=` and `user_tokenizer=`"
            " (pretrained weights cannot be downloaded in this offline environment)."
        )
    if isinstance(preds, str):
        preds = [preds]
    if isinstance(target, str):
        target = [target]
    model.eval()
    p_emb, p_mask = _embed(model, user_tokenizer, list(preds), max_length, device)
    t_emb, t_mask = _embed(model, user_tokenizer, list(target), max_length, device)
    precisions, recalls, f1s = [], [], []
    for i in range(len(preds)):
        pe = p_emb[i][p_mask[i]]
        te = t_emb[i][t_mask[i]]
        sim = pe @ te.t()
        precision = sim.max(dim=1).values.mean()
        recall = sim.max(dim=0).values.mean()
        f1 = 2 * precision * recall / (precision + recall + 1e-12)
        precisions.append(precision)
        recalls.append(recall)
        f1s.append(f1)
    out: Dict[str, Union[Tensor, List[float], str]] = {
        "precision": torch.stack(precisions),
        "recall": torch.stack(recalls),
        "f1": torch.stack(f1s),
    }
    if return_hash:
        out["hash"] = f"metrics_amd_bert_score(model={model_name_or_path or 'user'})"
    return out


@torch.no_grad()
def _mlm_dist(model, tokenizer, texts: List[str], temperature: float, max_length, device) -> Tensor:
    kwargs = {"return_tensors": "pt", "padding": True, "truncation": True}
    if max_length is not None:
        kwargs["max_length"] = max_length
    enc = tokenizer(texts, **kwargs)
    if device is not None:
        enc = {k: v.to(device) for k, v in enc.items()}
    out = model(**enc)
    logits = out.logits if hasattr(out, "logits") else out[0]
    probs = (logits / temperature).softmax(-1)
    mask = enc["attention_mask"].unsqueeze(-1)
    return (probs * mask).sum(1) / mask.sum(1)


def _information_measure(p: Tensor, t: Tensor, measure: str, alpha: Optional[float], beta: Optional[float]) -> Tensor:
    if measure == "kl_divergence":
        return (t * (t / p).log()).sum(-1)
    if measure == "alpha_divergence":
        a = alpha if alpha is not None else 0.5
        return (1 - (t**a * p ** (1 - a)).sum(-1)) / (a * (1 - a))
    if measure == "beta_divergence":
        b = beta if beta is not None else 0.5
        term = (t ** (b + 1)).sum(-1) / (b * (b + 1)) + (p ** (b + 1)).sum(-1) / (b + 1)
        return term - (t * p**b).sum(-1) / b
    if measure == "ab_divergence":
        a = alpha if alpha is not None else 0.5
        b = beta if beta is not None else 0.5
        out = (t ** (a + b)).sum(-1).log() / (b * (a + b))
        out = out + (p ** (a + b)).sum(-1).log() / (a * (a + b))
        return out - (t**a * p**b).sum(-1).log() / (a * b)
    if measure == "renyi_divergence":
        a = alpha if alpha is not None else 0.5
        return ((t**a * p ** (1 - a)).sum(-1)).log() / (a - 1)
    if measure == "l1_distance":
        return (t - p).abs().sum(-1)
    if measure == "l2_distance":
        return (t - p).pow(2).sum(-1).sqrt()
    if measure == "l_infinity_distance":
        return (t - p).abs().max(-1).values
    # fisher_rao_distance
    return 2 * torch.acos(((t * p).sqrt().sum(-1)).clamp(0, 1))


def infolm(
    preds: Union[str, Sequence[str]],
    target: Union[str, Sequence[str]],
    model_name_or_path: str = "bert-base-uncased",
    temperature: float = 0.25,
    information_measure: str = "kl_divergence",
    idf: bool = True,
    alpha: Optional[float] = None,
    beta: Optional[float] = None,
    device: Optional[Union[str, torch.device]] = None,
    max_length: Optional[int] = None,
    batch_size: int = 64,
    num_threads: int = 0,
    verbose: bool = True,
    return_sentence_level_score: bool = False,
    model: Optional[torch.nn.Module] = None,
    user_tokenizer: Any = None,
) -> Union[Tensor, Tuple[Tensor, Tensor]]:
    """InfoLM: information measure between masked-LM bag-of-token distributions."""
    if information_measure not in _ALLOWED_INFORMATION_MEASURE:
        raise ValueError(
            f"Argument `information_measure` expected one of {_ALLOWED_INFORMATION_MEASURE},"
            f" but got {information_measure}."
        )
    if model is None or user_tokenizer is None:
        raise ModuleNotFoundError(
            "`infolm` needs a local masked LM + tokenizer: pass `model=` and `user_tokenizer=`"
            " (pretrained weights cannot be downloaded in this offline environment)."
        )
    if isinstance(preds, str):
        preds = [preds]
    if isinstance(target, str):
        target = [target]
    model.eval()
    p = _mlm_dist(model, user_tokenizer, list(preds), temperature, max_length, device).clamp(min=1e-12)
    t = _mlm_dist(model, user_tokenizer, list(target), temperature, max_length, device).clamp(min=1e-12)
    scores = _information_measure(p, t, information_measure, alpha, beta)
    if return_sentence_level_score:
        return scores.mean(), scores
    return scores.mean()


__all__ = ["bert_score", "infolm"]
