"""DataFrame-only preprocessing (dask_ml Categorizer / DummyEncoder parity,
SURVEY §2.1 "dataframe-only" rows).

Operates on partitioned pandas DataFrames (core.bag.DistList of frames):
Categorizer learns the global category sets (distributed set-union, C12)
and casts object columns to pandas categoricals; DummyEncoder one-hot
expands categorical columns with the learned, globally consistent
category order.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import pandas as pd

from ..base import BaseEstimator, TransformerMixin, check_is_fitted
from ..core.bag import DistList


def _as_frames(X) -> DistList:
    if isinstance(X, DistList):
        return X
    if isinstance(X, pd.DataFrame):
        return DistList.from_global([X])
    raise TypeError("expected a pandas DataFrame or DistList of frames")


class Categorizer(BaseEstimator, TransformerMixin):
    def __init__(self, categories: Optional[Dict[str, List]] = None,
                 columns: Optional[List[str]] = None):
        self.categories = categories
        self.columns = columns

    def fit(self, X, y=None):
        frames = _as_frames(X)
        sample = frames.local[0] if frames.local else None
        cols_by_rank = frames.comm.all_gather_object(
            list(sample.columns) if sample is not None else None)
        cols = next(c for c in cols_by_rank if c is not None)
        if self.columns is not None:
            cat_cols = list(self.columns)
        else:
            dtypes_by_rank = frames.comm.all_gather_object(
                {c: str(sample[c].dtype) for c in cols}
                if sample is not None else None)
            dtypes = next(d for d in dtypes_by_rank if d is not None)
            cat_cols = [c for c in cols
                        if dtypes[c] in ("object", "category")]
        self.categories_ = {}
        for c in cat_cols:
            if self.categories and c in self.categories:
                cats = list(self.categories[c])
            else:
                cats = sorted(frames.fold_union(
                    lambda part, c=c: set(
                        v for f in part for v in f[c].dropna().unique())))
            self.categories_[c] = cats
        self.columns_ = cat_cols
        return self

    def transform(self, X, y=None):
        check_is_fitted(self, "categories_")
        frames = _as_frames(X)
        cats = self.categories_

        def _cast(df: pd.DataFrame) -> pd.DataFrame:
            out = df.copy()
            for c, values in cats.items():
                out[c] = pd.Categorical(out[c], categories=values)
            return out

        return frames.map(_cast)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class DummyEncoder(BaseEstimator, TransformerMixin):
    def __init__(self, columns: Optional[List[str]] = None,
                 drop_first: bool = False):
        self.columns = columns
        self.drop_first = drop_first

    def fit(self, X, y=None):
        frames = _as_frames(X)
        sample = frames.local[0] if frames.local else None
        info_by_rank = frames.comm.all_gather_object(
            {c: list(sample[c].cat.categories)
             for c in sample.columns
             if isinstance(sample[c].dtype, pd.CategoricalDtype)}
            if sample is not None else None)
        info = next(i for i in info_by_rank if i is not None)
        cols = self.columns if self.columns is not None else list(info)
        self.categories_ = {c: info[c] for c in cols}
        self.columns_ = cols
        return self

    def transform(self, X, y=None):
        check_is_fitted(self, "categories_")
        frames = _as_frames(X)
        cats = self.categories_
        drop_first = self.drop_first

        def _enc(df: pd.DataFrame) -> pd.DataFrame:
            out = df.copy()
            for c, values in cats.items():
                out[c] = pd.Categorical(out[c], categories=values)
            return pd.get_dummies(out, columns=list(cats),
                                  drop_first=drop_first)

        return frames.map(_enc)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)
