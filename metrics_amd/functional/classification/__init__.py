from metrics_amd.functional.classification.accuracy import (
    accuracy,
    binary_accuracy,
    multiclass_accuracy,
    multilabel_accuracy,
)
from metrics_amd.functional.classification.auroc import auroc, binary_auroc, multiclass_auroc, multilabel_auroc
from metrics_amd.functional.classification.average_precision import (
    average_precision,
    binary_average_precision,
    multiclass_average_precision,
    multilabel_average_precision,
)
from metrics_amd.functional.classification.calibration_error import (
    binary_calibration_error,
    calibration_error,
    multiclass_calibration_error,
)
from metrics_amd.functional.classification.cohen_kappa import binary_cohen_kappa, cohen_kappa, multiclass_cohen_kappa
from metrics_amd.functional.classification.confusion_matrix import (
    binary_confusion_matrix,
    confusion_matrix,
    multiclass_confusion_matrix,
    multilabel_confusion_matrix,
)
from metrics_amd.functional.classification.exact_match import (
    exact_match,
    multiclass_exact_match,
    multilabel_exact_match,
)
from metrics_amd.functional.classification.f_beta import (
    binary_f1_score,
    binary_fbeta_score,
    f1_score,
    fbeta_score,
    multiclass_f1_score,
    multiclass_fbeta_score,
    multilabel_f1_score,
    multilabel_fbeta_score,
)
from metrics_amd.functional.classification.hamming import (
    binary_hamming_distance,
    hamming_distance,
    multiclass_hamming_distance,
    multilabel_hamming_distance,
)
from metrics_amd.functional.classification.hinge import binary_hinge_loss, hinge_loss, multiclass_hinge_loss
from metrics_amd.functional.classification.jaccard import (
    binary_jaccard_index,
    jaccard_index,
    multiclass_jaccard_index,
    multilabel_jaccard_index,
)
from metrics_amd.functional.classification.matthews_corrcoef import (
    binary_matthews_corrcoef,
    matthews_corrcoef,
    multiclass_matthews_corrcoef,
    multilabel_matthews_corrcoef,
)
from metrics_amd.functional.classification.negative_predictive_value import (
    binary_negative_predictive_value,
    multiclass_negative_predictive_value,
    multilabel_negative_predictive_value,
    negative_predictive_value,
)
from metrics_amd.functional.classification.precision_recall import (
    binary_precision,
    binary_recall,
    multiclass_precision,
    multiclass_recall,
    multilabel_precision,
    multilabel_recall,
    precision,
    recall,
)
from metrics_amd.functional.classification.precision_recall_curve import (
    binary_precision_recall_curve,
    multiclass_precision_recall_curve,
    multilabel_precision_recall_curve,
    precision_recall_curve,
)
from metrics_amd.functional.classification.ranking import (
    multilabel_coverage_error,
    multilabel_ranking_average_precision,
    multilabel_ranking_loss,
)
from metrics_amd.functional.classification.roc import binary_roc, multiclass_roc, multilabel_roc, roc
from metrics_amd.functional.classification.specificity import (
    binary_specificity,
    multiclass_specificity,
    multilabel_specificity,
    specificity,
)
from metrics_amd.functional.classification.stat_scores import (
    binary_stat_scores,
    multiclass_stat_scores,
    multilabel_stat_scores,
    stat_scores,
)
from metrics_amd.functional.classification.at_fixed import (
    binary_logauc,
    binary_precision_at_fixed_recall,
    binary_recall_at_fixed_precision,
    binary_sensitivity_at_specificity,
    binary_specificity_at_sensitivity,
    multiclass_logauc,
    multiclass_precision_at_fixed_recall,
    multiclass_recall_at_fixed_precision,
    multiclass_sensitivity_at_specificity,
    multiclass_specificity_at_sensitivity,
    multilabel_logauc,
    multilabel_precision_at_fixed_recall,
    multilabel_recall_at_fixed_precision,
    multilabel_sensitivity_at_specificity,
    multilabel_specificity_at_sensitivity,
    logauc,
    precision_at_fixed_recall,
    recall_at_fixed_precision,
    sensitivity_at_specificity,
    specificity_at_sensitivity,
)
from metrics_amd.functional.classification.group_fairness import (
    binary_fairness,
    binary_groups_stat_rates,
    demographic_parity,
    equal_opportunity,
)
from metrics_amd.functional.classification.dice import dice
from metrics_amd.functional.segmentation import generalized_dice_score  # noqa: F401  (reference re-export)
