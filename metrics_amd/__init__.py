"""metrics_amd — an MI355X-native metrics engine with the torchmetrics API.

Compute path: PyTorch-ROCm tensors + in-tree gfx950 HIP kernels for the hot
reductions (see ``csrc/``); distributed state sync over RCCL/xGMI with the
reductions mapped onto fused all-reduces (see ``utilities/distributed.py``).
"""
import logging as __logging
import os

_logger = __logging.getLogger("metrics_amd")
_logger.addHandler(__logging.StreamHandler())
_logger.setLevel(__logging.INFO)

_PACKAGE_ROOT = os.path.dirname(__file__)

__version__ = "0.1.0"

from metrics_amd.aggregation import (  # noqa: E402
    CatMetric,
    MaxMetric,
    MeanMetric,
    MinMetric,
    RunningMean,
    RunningSum,
    SumMetric,
)
from metrics_amd.collections import MetricCollection  # noqa: E402
from metrics_amd.metric import CompositionalMetric, Metric  # noqa: E402
from metrics_amd import audio, classification, clustering, detection, functional, image, multimodal, nominal, ops, regression, retrieval, segmentation, shape, text, utilities, wrappers  # noqa: E402
from metrics_amd.classification import *  # noqa: E402,F401,F403
from metrics_amd.regression import *  # noqa: E402,F401,F403
from metrics_amd.retrieval import *  # noqa: E402,F401,F403
from metrics_amd.clustering import *  # noqa: E402,F401,F403
from metrics_amd.nominal import *  # noqa: E402,F401,F403
from metrics_amd.detection import *  # noqa: E402,F401,F403
from metrics_amd.segmentation import *  # noqa: E402,F401,F403
from metrics_amd.image import *  # noqa: E402,F401,F403
from metrics_amd.audio import *  # noqa: E402,F401,F403
from metrics_amd.text import *  # noqa: E402,F401,F403
from metrics_amd.shape import ProcrustesDisparity  # noqa: E402,F401
from metrics_amd.wrappers import (  # noqa: E402,F401
    BinaryTargetTransformer,
    BootStrapper,
    ClasswiseWrapper,
    FeatureShare,
    LambdaInputTransformer,
    MetricTracker,
    MinMaxMetric,
    MultioutputWrapper,
    MultitaskWrapper,
    Running,
)

__all__ = [
    "CatMetric",
    "CompositionalMetric",
    "MaxMetric",
    "MeanMetric",
    "Metric",
    "MetricCollection",
    "MinMetric",
    "RunningMean",
    "RunningSum",
    "SumMetric",
    "classification",
    "regression",
    "functional",
    "ops",
    "utilities",
    "wrappers",
]
__all__ += classification.__all__
__all__ += regression.__all__
__all__ += retrieval.__all__
__all__ += clustering.__all__
__all__ += nominal.__all__
__all__ += detection.__all__
__all__ += segmentation.__all__
__all__ += image.__all__
__all__ += audio.__all__
__all__ += text.__all__
__all__ += ["ProcrustesDisparity"]
__all__ += wrappers.__all__

from metrics_amd import graphs  # noqa: F401,E402  (hipGraph-captured updates)
