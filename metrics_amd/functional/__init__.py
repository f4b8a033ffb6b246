"""Functional metric API (L3). Parity: torchmetrics ``functional/__init__.py``."""
from metrics_amd.functional.classification import *  # noqa: F401,F403
from metrics_amd.functional.regression import *  # noqa: F401,F403
from metrics_amd.functional.retrieval import *  # noqa: F401,F403
from metrics_amd.functional.clustering import *  # noqa: F401,F403
from metrics_amd.functional.nominal import *  # noqa: F401,F403
from metrics_amd.functional.detection import *  # noqa: F401,F403
from metrics_amd.functional.segmentation import *  # noqa: F401,F403
from metrics_amd.functional.image import *  # noqa: F401,F403
from metrics_amd.functional.audio import *  # noqa: F401,F403
from metrics_amd.functional.text import *  # noqa: F401,F403
from metrics_amd.functional.shape import procrustes_disparity  # noqa: F401
from metrics_amd.functional.multimodal import clip_image_quality_assessment, clip_score  # noqa: F401
from metrics_amd.functional.pairwise import (  # noqa: F401
    pairwise_cosine_similarity,
    pairwise_euclidean_distance,
    pairwise_linear_similarity,
    pairwise_manhattan_distance,
    pairwise_minkowski_distance,
)
from metrics_amd.functional import audio, classification, clustering, detection, image, multimodal, nominal, pairwise, regression, retrieval, segmentation, shape, text  # noqa: F401
