from metrics_amd.regression.concordance import ConcordanceCorrCoef
from metrics_amd.regression.cosine_similarity import CosineSimilarity
from metrics_amd.regression.csi import CriticalSuccessIndex
from metrics_amd.regression.explained_variance import ExplainedVariance
from metrics_amd.regression.kl_divergence import KLDivergence
from metrics_amd.regression.log_cosh import LogCoshError
from metrics_amd.regression.log_mse import MeanSquaredLogError
from metrics_amd.regression.mae import MeanAbsoluteError
from metrics_amd.regression.mape import MeanAbsolutePercentageError
from metrics_amd.regression.minkowski import MinkowskiDistance
from metrics_amd.regression.mse import MeanSquaredError
from metrics_amd.regression.nrmse import NormalizedRootMeanSquaredError
from metrics_amd.regression.pearson import PearsonCorrCoef
from metrics_amd.regression.r2 import R2Score
from metrics_amd.regression.rse import RelativeSquaredError
from metrics_amd.regression.spearman import KendallRankCorrCoef, SpearmanCorrCoef
from metrics_amd.regression.symmetric_mape import SymmetricMeanAbsolutePercentageError
from metrics_amd.regression.tweedie_deviance import TweedieDevianceScore
from metrics_amd.regression.weighted_mape import WeightedMeanAbsolutePercentageError

__all__ = [
    "ConcordanceCorrCoef",
    "CosineSimilarity",
    "CriticalSuccessIndex",
    "ExplainedVariance",
    "KLDivergence",
    "KendallRankCorrCoef",
    "LogCoshError",
    "MeanAbsoluteError",
    "MeanAbsolutePercentageError",
    "MeanSquaredError",
    "MeanSquaredLogError",
    "MinkowskiDistance",
    "NormalizedRootMeanSquaredError",
    "PearsonCorrCoef",
    "R2Score",
    "RelativeSquaredError",
    "SpearmanCorrCoef",
    "SymmetricMeanAbsolutePercentageError",
    "TweedieDevianceScore",
    "WeightedMeanAbsolutePercentageError",
]
