from metrics_amd.utilities.checks import _check_same_shape, check_forward_full_state_property
from metrics_amd.utilities.data import (
    apply_to_collection,
    dim_zero_cat,
    dim_zero_max,
    dim_zero_mean,
    dim_zero_min,
    dim_zero_sum,
)
from metrics_amd.utilities.distributed import class_reduce, reduce
from metrics_amd.utilities.prints import rank_zero_debug, rank_zero_info, rank_zero_warn

__all__ = [
    "_check_same_shape",
    "apply_to_collection",
    "check_forward_full_state_property",
    "class_reduce",
    "dim_zero_cat",
    "dim_zero_max",
    "dim_zero_mean",
    "dim_zero_min",
    "dim_zero_sum",
    "rank_zero_debug",
    "rank_zero_info",
    "rank_zero_warn",
    "reduce",
]
