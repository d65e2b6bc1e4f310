"""xgboost / lightgbm hand-off (dask_ml/xgboost.py parity).

The reference pattern: start the native distributed trainer's tracker next
to the workers and hand each worker its local partitions in place
(SURVEY §2.1). The MI355X equivalent hands each rank's local shard to
xgboost's collective (rabit/federated) training. xgboost is not installed
in this environment; the API surface raises an informative error until it
is.
"""

from __future__ import annotations

from .base import check_X_y


def _require_xgboost():
    try:
        import xgboost  # noqa: F401
        return xgboost
    except ImportError as e:  # pragma: no cover
        raise ImportError(
            "mi355ml.xgboost requires the `xgboost` package (with its "
            "collective/rabit support) which is not installed in this "
            "environment. Each rank would pass its local shard to "
            "xgb.train inside a CollectiveCommunicator context."
        ) from e


def train(params, X, y, num_boost_round: int = 10, **kwargs):
    xgb = _require_xgboost()
    X, y = check_X_y(X, y, dtype=None)
    dtrain = xgb.DMatrix(X.local.cpu().numpy(), label=y.local.cpu().numpy())
    # one rabit worker per rank; xgboost allreduces histograms internally
    with xgb.collective.CommunicatorContext(
            dmlc_communicator="rabit"):  # pragma: no cover
        return xgb.train(params, dtrain, num_boost_round=num_boost_round,
                         **kwargs)


def predict(model, X):
    xgb = _require_xgboost()
    from .base import check_array
    from .core.array import DistArray
    import torch

    X = check_array(X, dtype=None)

    def _p(t):
        d = xgb.DMatrix(t.cpu().numpy())
        return torch.as_tensor(model.predict(d), device=t.device)

    return X.map_local(_p)
