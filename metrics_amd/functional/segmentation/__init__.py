from metrics_amd.functional.segmentation.metrics import (
    dice_score,
    generalized_dice_score,
    hausdorff_distance,
    mean_iou,
)
