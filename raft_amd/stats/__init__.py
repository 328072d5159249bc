"""Statistics & metrics (reference: raft/stats/*)."""
from .moments import sum_cols, mean, stddev, vars_, meanvar, minmax, weighted_mean, mean_center, mean_add
from .cov import cov
from .histogram import histogram
from .contingency import contingency_matrix
from .clustering import (
    adjusted_rand_index, rand_index, mutual_info_score, entropy,
    homogeneity_score, completeness_score, v_measure, kl_divergence,
    dispersion, silhouette_score, silhouette_score_batched,
)
from .regression import r2_score, regression_metrics, information_criterion
from .classification import accuracy_score
from .neighborhood import neighborhood_recall, trustworthiness_score

__all__ = [
    "sum_cols", "mean", "stddev", "vars_", "meanvar", "minmax", "weighted_mean",
    "mean_center", "mean_add", "cov", "histogram", "contingency_matrix",
    "adjusted_rand_index", "rand_index", "mutual_info_score", "entropy",
    "homogeneity_score", "completeness_score", "v_measure", "kl_divergence",
    "dispersion", "silhouette_score", "silhouette_score_batched", "r2_score", "regression_metrics",
    "information_criterion", "accuracy_score", "neighborhood_recall",
    "trustworthiness_score",
]
