"""Functional multimodal metrics (CLIP-backed).

Parity: reference functional/multimodal/{clip_score,clip_iqa}.py. The
reference downloads CLIP checkpoints from HuggingFace; offline, both
functions accept an explicit ``model``/``processor`` pair and raise
``ModuleNotFoundError`` otherwise (matching the reference's behavior when
``transformers`` is missing).
"""
from metrics_amd.functional.multimodal.clip import clip_image_quality_assessment, clip_score

__all__ = ["clip_image_quality_assessment", "clip_score"]
