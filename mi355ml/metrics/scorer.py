"""Scorer registry (dask_ml/metrics/scorer.py parity): SCORERS dict,
get_scorer, check_scoring."""

from __future__ import annotations

from typing import Callable, Union

from .classification import accuracy_score, log_loss
from .regression import (mean_absolute_error, mean_squared_error, r2_score)


def _accuracy(est, X, y):
    return accuracy_score(y, est.predict(X))


def _neg_mse(est, X, y):
    return -mean_squared_error(y, est.predict(X))


def _neg_rmse(est, X, y):
    return -mean_squared_error(y, est.predict(X), squared=False)


def _neg_mae(est, X, y):
    return -mean_absolute_error(y, est.predict(X))


def _r2(est, X, y):
    return r2_score(y, est.predict(X))


def _neg_log_loss(est, X, y):
    proba = est.predict_proba(X)
    p = proba.map_local(lambda t: t[:, 1]) if proba.ncols == 2 else proba
    return -log_loss(y, p)


SCORERS = {
    "accuracy": _accuracy,
    "neg_mean_squared_error": _neg_mse,
    "neg_root_mean_squared_error": _neg_rmse,
    "neg_mean_absolute_error": _neg_mae,
    "r2": _r2,
    "neg_log_loss": _neg_log_loss,
}


def get_scorer(scoring: Union[str, Callable]) -> Callable:
    if callable(scoring):
        return scoring
    try:
        return SCORERS[scoring]
    except KeyError:
        raise ValueError(
            f"unknown scorer {scoring!r}; available: {sorted(SCORERS)}")


def check_scoring(estimator, scoring=None) -> Callable:
    if scoring is None:
        def _default(est, X, y=None):
            return est.score(X, y) if y is not None else est.score(X)
        return _default
    return get_scorer(scoring)
