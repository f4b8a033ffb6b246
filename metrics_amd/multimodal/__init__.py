"""Multimodal metrics. Parity: torchmetrics ``multimodal/{clip_score,clip_iqa}.py``.

Both wrap a CLIP-style model; offline, pass local ``model=`` + ``processor=``.
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence, Union

import torch
from torch import Tensor

from metrics_amd.metric import Metric


class CLIPScore(Metric):
    """CLIPScore: 100 * max(cos(image_emb, text_emb), 0), averaged.

    Requires a local CLIP model + processor (HuggingFace-style API:
    ``model.get_image_features`` / ``model.get_text_features``).
    """

    is_differentiable = False
    higher_is_better = True
    full_state_update = True
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 100.0

    score: Tensor
    n_samples: Tensor

    def __init__(
        self,
        model_name_or_path: Optional[str] = None,
        model=None,
        processor=None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if model is None or processor is None:
            raise ModuleNotFoundError(
                "CLIPScore needs a local CLIP model + processor: pass `model=` and `processor=`"
                " (weights cannot be downloaded in this offline environment)."
            )
        self.model = model
        self.processor = processor
        self.model.eval()
        self.add_state("score", torch.tensor(0.0), dist_reduce_fx="sum")
        self.add_state("n_samples", torch.tensor(0, dtype=torch.long), dist_reduce_fx="sum")

    @torch.no_grad()
    def update(self, images: Union[Tensor, List[Tensor]], text: Union[str, List[str]]) -> None:
        """Accumulate image-text similarity scores."""
        if isinstance(text, str):
            text = [text]
        if isinstance(images, Tensor) and images.ndim == 3:
            images = [images]
        img_list = list(images) if not isinstance(images, Tensor) else list(images)
        if len(text) != len(img_list):
            raise ValueError(
                f"Expected the number of images and text examples to be the same but got {len(img_list)} and {len(text)}"
            )
        processed = self.processor(text=text, images=[i.cpu() for i in img_list], return_tensors="pt", padding=True)
        img_features = self.model.get_image_features(processed["pixel_values"].to(self.device))
        img_features = img_features / img_features.norm(p=2, dim=-1, keepdim=True)
        txt_features = self.model.get_text_features(
            processed["input_ids"].to(self.device), processed["attention_mask"].to(self.device)
        )
        txt_features = txt_features / txt_features.norm(p=2, dim=-1, keepdim=True)
        score = 100 * (img_features * txt_features).sum(axis=-1)
        self.score += score.sum(0)
        self.n_samples += img_features.shape[0]

    def compute(self) -> Tensor:
        """Average CLIP score (clamped at 0)."""
        return torch.max(self.score / self.n_samples, torch.zeros_like(self.score))

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


class CLIPImageQualityAssessment(Metric):
    """CLIP-IQA: softmax over positive/negative prompt similarities.

    Requires a local CLIP model + processor and prompt pairs.
    """

    is_differentiable = False
    higher_is_better = True
    full_state_update = True
    plot_lower_bound: float = 0.0
    plot_upper_bound: float = 1.0

    _PROMPTS = {
        "quality": ("Good photo.", "Bad photo."),
        "brightness": ("Bright photo.", "Dark photo."),
        "noisiness": ("Clean photo.", "Noisy photo."),
        "colorfullness": ("Colorful photo.", "Dull photo."),
        "sharpness": ("Sharp photo.", "Blurry photo."),
        "contrast": ("High contrast photo.", "Low contrast photo."),
        "complexity": ("Complex photo.", "Simple photo."),
        "natural": ("Natural photo.", "Synthetic photo."),
        "happy": ("Happy photo.", "Sad photo."),
        "scary": ("Scary photo.", "Peaceful photo."),
        "new": ("New photo.", "Old photo."),
        "warm": ("Warm photo.", "Cold photo."),
        "real": ("Real photo.", "Abstract photo."),
        "beautiful": ("Beautiful photo.", "Ugly photo."),
        "lonely": ("Lonely photo.", "Sociable photo."),
        "relaxing": ("Relaxing photo.", "Stressful photo."),
    }

    def __init__(
        self,
        model_name_or_path: Optional[str] = None,
        data_range: float = 1.0,
        prompts: tuple = ("quality",),
        model=None,
        processor=None,
        **kwargs: Any,
    ) -> None:
        super().__init__(**kwargs)
        if model is None or processor is None:
            raise ModuleNotFoundError(
                "CLIPImageQualityAssessment needs a local CLIP model + processor: pass `model=` and `processor=`."
            )
        self.model = model
        self.processor = processor
        self.model.eval()
        self.data_range = data_range
        prompt_pairs = []
        names = []
        for p in prompts:
            if isinstance(p, str):
                if p not in self._PROMPTS:
                    raise ValueError(f"Unknown prompt {p}; expected one of {list(self._PROMPTS)} or a (pos, neg) tuple")
                prompt_pairs.append(self._PROMPTS[p])
                names.append(p)
            elif isinstance(p, tuple) and len(p) == 2:
                prompt_pairs.append(p)
                names.append(p[0])
            else:
                raise ValueError("Prompts must be strings or (positive, negative) tuples")
        self.prompt_pairs = prompt_pairs
        self.prompt_names = names
        self.add_state("probs", [], dist_reduce_fx="cat")

    @torch.no_grad()
    def update(self, images: Tensor) -> None:
        """Accumulate per-image, per-prompt probabilities."""
        texts = [t for pair in self.prompt_pairs for t in pair]
        processed = self.processor(text=texts, images=[i.cpu() for i in images], return_tensors="pt", padding=True)
        img_features = self.model.get_image_features(processed["pixel_values"].to(self.device))
        img_features = img_features / img_features.norm(p=2, dim=-1, keepdim=True)
        txt_features = self.model.get_text_features(
            processed["input_ids"].to(self.device), processed["attention_mask"].to(self.device)
        )
        txt_features = txt_features / txt_features.norm(p=2, dim=-1, keepdim=True)
        logits = 100 * img_features @ txt_features.t()  # (N, 2*P)
        logits = logits.reshape(img_features.shape[0], len(self.prompt_pairs), 2)
        probs = logits.softmax(-1)[..., 0]
        self.probs.append(probs)

    def compute(self):
        """Mean probability per prompt (dict when multiple prompts)."""
        from metrics_amd.utilities.data import dim_zero_cat

        probs = dim_zero_cat(self.probs)
        if len(self.prompt_names) == 1:
            return probs.squeeze(-1)
        return {name: probs[:, i] for i, name in enumerate(self.prompt_names)}

    def plot(self, val=None, ax=None):
        return self._plot(val, ax)


__all__ = ["CLIPImageQualityAssessment", "CLIPScore"]
