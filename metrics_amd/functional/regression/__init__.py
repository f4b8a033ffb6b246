from metrics_amd.functional.regression.concordance import concordance_corrcoef
from metrics_amd.functional.regression.cosine_similarity import cosine_similarity
from metrics_amd.functional.regression.csi import critical_success_index
from metrics_amd.functional.regression.explained_variance import explained_variance
from metrics_amd.functional.regression.kendall import kendall_rank_corrcoef
from metrics_amd.functional.regression.kl_divergence import kl_divergence
from metrics_amd.functional.regression.log_cosh import log_cosh_error
from metrics_amd.functional.regression.log_mse import mean_squared_log_error
from metrics_amd.functional.regression.mae import mean_absolute_error
from metrics_amd.functional.regression.mape import (
    mean_absolute_percentage_error,
    symmetric_mean_absolute_percentage_error,
    weighted_mean_absolute_percentage_error,
)
from metrics_amd.functional.regression.minkowski import minkowski_distance
from metrics_amd.functional.regression.mse import mean_squared_error
from metrics_amd.functional.regression.nrmse import normalized_root_mean_squared_error
from metrics_amd.functional.regression.pearson import pearson_corrcoef
from metrics_amd.functional.regression.r2 import r2_score
from metrics_amd.functional.regression.rse import relative_squared_error
from metrics_amd.functional.regression.spearman import spearman_corrcoef
from metrics_amd.functional.regression.tweedie_deviance import tweedie_deviance_score
