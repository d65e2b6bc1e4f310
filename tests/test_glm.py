"""GLM estimators/solvers vs sklearn oracle (tolerance-based)."""

import numpy as np
import pytest
from sklearn.linear_model import LinearRegression as SkLin
from sklearn.linear_model import LogisticRegression as SkLog

from mi355ml.core.array import from_numpy
from mi355ml.linear_model import (LinearRegression, LogisticRegression,
                                  PoissonRegression)


def _cls_data(n=4000, f=8, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f)
    p = 1 / (1 + np.exp(-(X @ w + 0.3)))
    y = (rng.uniform(size=n) < p).astype(np.float32)
    return X, y


def _reg_data(n=3000, f=6, seed=1):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f) * 3
    y = (X @ w + 1.5 + rng.standard_normal(n) * 0.01).astype(np.float32)
    return X, y, w


def test_logreg_lbfgs_matches_sklearn():
    X, y = _cls_data()
    ours = LogisticRegression(C=1.0, solver="lbfgs", max_iter=200).fit(
        from_numpy(X), from_numpy(y))
    ref = SkLog(C=1.0, solver="lbfgs", max_iter=500).fit(X, y)
    np.testing.assert_allclose(ours.coef_.cpu().numpy(), ref.coef_[0], rtol=0.05,
                               atol=0.02)
    np.testing.assert_allclose(ours.intercept_, ref.intercept_[0], rtol=0.05,
                               atol=0.02)
    acc_ours = ours.score(from_numpy(X), from_numpy(y))
    acc_ref = ref.score(X, y)
    assert acc_ours >= acc_ref - 0.01


@pytest.mark.parametrize("solver", ["gradient_descent", "newton", "admm",
                                    "proximal_grad"])
def test_logreg_other_solvers(solver):
    X, y = _cls_data(n=2000, f=5, seed=2)
    kw = dict(C=1e3, solver=solver, max_iter=200)
    if solver == "admm":
        kw["max_iter"] = 30
    ours = LogisticRegression(**kw).fit(from_numpy(X), from_numpy(y))
    ref = SkLog(C=1e3, max_iter=1000).fit(X, y)
    acc_ours = ours.score(from_numpy(X), from_numpy(y))
    assert acc_ours >= ref.score(X, y) - 0.03, (solver, acc_ours)


def test_logreg_l1_sparsity():
    X, y = _cls_data(n=2000, f=10, seed=3)
    # only 10 informative-ish dims anyway; strong l1 should zero some coords
    ours = LogisticRegression(C=0.001, penalty="l1", solver="admm",
                              max_iter=30).fit(from_numpy(X), from_numpy(y))
    assert (ours.coef_.abs() < 1e-6).sum() > 0


def test_linreg_matches_sklearn():
    X, y, w = _reg_data()
    ours = LinearRegression(C=1e8, solver="lbfgs", max_iter=500).fit(
        from_numpy(X), from_numpy(y))
    ref = SkLin().fit(X, y)
    np.testing.assert_allclose(ours.coef_.cpu().numpy(), ref.coef_, rtol=1e-2,
                               atol=1e-3)
    assert ours.score(from_numpy(X), from_numpy(y)) > 0.999


def test_poisson_regression_recovers():
    rng = np.random.default_rng(4)
    X = rng.standard_normal((3000, 4)).astype(np.float32)
    w = np.array([0.4, -0.3, 0.2, 0.0])
    lam = np.exp(X @ w + 0.5)
    y = rng.poisson(lam).astype(np.float32)
    ours = PoissonRegression(C=1e6, solver="lbfgs", max_iter=300).fit(
        from_numpy(X), from_numpy(y))
    np.testing.assert_allclose(ours.coef_.cpu().numpy(), w, atol=0.05)
    np.testing.assert_allclose(ours.intercept_, 0.5, atol=0.05)


def _logreg_world2(rank, world, X_bytes, y_bytes, n, f, coef_ref):
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    ours = LogisticRegression(C=1.0, solver="lbfgs", max_iter=200).fit(
        from_numpy(X), from_numpy(y))
    np.testing.assert_allclose(ours.coef_.cpu().numpy(), coef_ref, rtol=0.05,
                               atol=0.03)


def test_logreg_world2(mp_runner):
    X, y = _cls_data(n=2001, f=5, seed=6)
    ref = SkLog(C=1.0, max_iter=500).fit(X, y)
    mp_runner(_logreg_world2, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1],
                    ref.coef_[0]))


def test_poisson_partial_fit_improves():
    rng = np.random.default_rng(23)
    X = rng.standard_normal((4000, 5)).astype(np.float32)
    w = np.array([0.4, -0.3, 0.2, 0.0, 0.1])
    y = rng.poisson(np.exp(X @ w)).astype(np.float32)
    pr = PoissonRegression(eta0=0.05)
    from mi355ml.core.array import from_numpy as fn
    losses = []
    from mi355ml.glm.algorithms import _dist_value_grad
    import torch
    for ep in range(30):
        pr.partial_fit(fn(X), fn(y))
        l, _, _ = _dist_value_grad(fn(X), fn(y),
                                   pr.coef_, pr.intercept_, "poisson")
        losses.append(l)
    assert losses[-1] < losses[0] - 1.0  # SGD steps reduce the deviance
    np.testing.assert_allclose(pr.coef_.cpu().numpy(), w, atol=0.25)
