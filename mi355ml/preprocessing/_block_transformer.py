"""BlockTransformer (dask_ml/preprocessing/_block_transformer.py parity):
apply a stateless user function to every shard."""

from __future__ import annotations

from ..base import BaseEstimator, TransformerMixin, check_array


class BlockTransformer(BaseEstimator, TransformerMixin):
    def __init__(self, func, validate: bool = False, **kw_args):
        self.func = func
        self.validate = validate
        self.kw_args = kw_args

    def fit(self, X, y=None):
        return self

    def transform(self, X):
        X = check_array(X, dtype=None)
        kw = self.kw_args or {}
        return X.map_local(lambda t: self.func(t, **kw))

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)
