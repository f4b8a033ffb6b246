"""Functional CLIPScore / CLIP-IQA via user-supplied CLIP models."""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Union

import torch
from torch import Tensor


def clip_score(
    source: Union[Tensor, List[Tensor], List[str], str],
    target: Union[Tensor, List[Tensor], List[str], str],
    model_name_or_path: str = "openai/clip-vit-large-patch14",
    model: Any = None,
    processor: Any = None,
) -> Tensor:
    """CLIPScore(images, text) = 100 * max(cos(img_emb, txt_emb), 0), averaged.

    Requires a local CLIP ``model`` + ``processor`` (HuggingFace API).
    """
    from metrics_amd.multimodal import CLIPScore

    metric = CLIPScore(model_name_or_path=model_name_or_path, model=model, processor=processor)
    metric.update(source, target)
    return metric.compute()


def clip_image_quality_assessment(
    images: Tensor,
    model_name_or_path: str = "clip_iqa",
    data_range: float = 1.0,
    prompts: tuple = ("quality",),
    model: Any = None,
    processor: Any = None,
) -> Union[Tensor, Dict[str, Tensor]]:
    """CLIP-IQA: softmax over positive/negative prompt similarity per image."""
    from metrics_amd.multimodal import CLIPImageQualityAssessment

    metric = CLIPImageQualityAssessment(
        model_name_or_path=model_name_or_path, data_range=data_range, prompts=prompts,
        model=model, processor=processor,
    )
    metric.update(images)
    return metric.compute()
