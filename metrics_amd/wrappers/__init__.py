from metrics_amd.wrappers.abstract import WrapperMetric
from metrics_amd.wrappers.running import Running

__all__ = ["Running", "WrapperMetric"]
