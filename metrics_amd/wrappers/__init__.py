from metrics_amd.wrappers.abstract import WrapperMetric
from metrics_amd.wrappers.bootstrapping import BootStrapper
from metrics_amd.wrappers.classwise import ClasswiseWrapper
from metrics_amd.wrappers.feature_share import FeatureShare
from metrics_amd.wrappers.minmax import MinMaxMetric
from metrics_amd.wrappers.multioutput import MultioutputWrapper
from metrics_amd.wrappers.multitask import MultitaskWrapper
from metrics_amd.wrappers.running import Running
from metrics_amd.wrappers.tracker import MetricTracker
from metrics_amd.wrappers.transformations import BinaryTargetTransformer, LambdaInputTransformer, MetricInputTransformer

__all__ = [
    "BinaryTargetTransformer",
    "BootStrapper",
    "ClasswiseWrapper",
    "FeatureShare",
    "LambdaInputTransformer",
    "MetricInputTransformer",
    "MetricTracker",
    "MinMaxMetric",
    "MultioutputWrapper",
    "MultitaskWrapper",
    "Running",
    "WrapperMetric",
]
