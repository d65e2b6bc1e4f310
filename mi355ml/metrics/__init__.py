from .regression import mean_squared_error, mean_absolute_error, r2_score
from .classification import accuracy_score, log_loss
from .pairwise import (euclidean_distances, pairwise_distances_argmin_min,
                       pairwise_distances, linear_kernel, rbf_kernel,
                       polynomial_kernel, sigmoid_kernel)
from .scorer import SCORERS, check_scoring, get_scorer

__all__ = [
    "mean_squared_error", "mean_absolute_error", "r2_score",
    "accuracy_score", "log_loss",
    "euclidean_distances", "pairwise_distances_argmin_min",
    "pairwise_distances", "linear_kernel", "rbf_kernel",
    "polynomial_kernel", "sigmoid_kernel",
    "SCORERS", "check_scoring", "get_scorer",
]
