from metrics_amd.functional.detection.iou import (
    complete_intersection_over_union,
    distance_intersection_over_union,
    generalized_intersection_over_union,
    intersection_over_union,
)
from metrics_amd.detection.panoptic_quality import modified_panoptic_quality, panoptic_quality
