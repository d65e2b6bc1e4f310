"""Text feature extraction (dask_ml/feature_extraction/text.py parity).

HashingVectorizer / FeatureHasher: stateless per-partition sklearn
transforms producing a DistList of scipy-sparse blocks (the reference's
array-of-sparse-blocks with unknown row chunks). CountVectorizer: the
vocabulary is a distributed set-union over partitions (SURVEY C12 — the
reference's bag foldby), then a per-partition transform with the shared
vocabulary.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
from sklearn.base import clone
from sklearn.feature_extraction.text import CountVectorizer as _SkCount
from sklearn.feature_extraction.text import FeatureHasher as _SkHasher
from sklearn.feature_extraction.text import HashingVectorizer as _SkHash

from ..base import BaseEstimator, TransformerMixin, check_is_fitted
from ..core.bag import DistList


def _as_distlist(docs) -> DistList:
    if isinstance(docs, DistList):
        return docs
    return DistList.from_global(list(docs))


class HashingVectorizer(BaseEstimator, TransformerMixin):
    def __init__(self, n_features=1 << 20, norm="l2", alternate_sign=True,
                 binary=False, lowercase=True, ngram_range=(1, 1)):
        self.n_features = n_features
        self.norm = norm
        self.alternate_sign = alternate_sign
        self.binary = binary
        self.lowercase = lowercase
        self.ngram_range = ngram_range

    def _vec(self):
        return _SkHash(n_features=self.n_features, norm=self.norm,
                       alternate_sign=self.alternate_sign,
                       binary=self.binary, lowercase=self.lowercase,
                       ngram_range=self.ngram_range)

    def fit(self, X, y=None):
        return self

    def transform(self, X):
        docs = _as_distlist(X)
        vec = self._vec()
        return docs.map_partition(
            lambda part: [vec.transform(part)] if part else [])

    def fit_transform(self, X, y=None):
        return self.transform(X)


class FeatureHasher(BaseEstimator, TransformerMixin):
    def __init__(self, n_features=1 << 20, input_type="dict",
                 alternate_sign=True):
        self.n_features = n_features
        self.input_type = input_type
        self.alternate_sign = alternate_sign

    def fit(self, X, y=None):
        return self

    def transform(self, X):
        items = _as_distlist(X)
        h = _SkHasher(n_features=self.n_features, input_type=self.input_type,
                      alternate_sign=self.alternate_sign)
        return items.map_partition(
            lambda part: [h.transform(part)] if part else [])

    def fit_transform(self, X, y=None):
        return self.transform(X)


class CountVectorizer(BaseEstimator, TransformerMixin):
    def __init__(self, lowercase=True, ngram_range=(1, 1), binary=False,
                 vocabulary=None):
        self.lowercase = lowercase
        self.ngram_range = ngram_range
        self.binary = binary
        self.vocabulary = vocabulary

    def fit(self, X, y=None):
        docs = _as_distlist(X)
        if self.vocabulary is not None:
            vocab = sorted(self.vocabulary)
        else:
            def _part_vocab(part):
                if not part:
                    return set()
                v = _SkCount(lowercase=self.lowercase,
                             ngram_range=self.ngram_range)
                v.fit(part)
                return set(v.vocabulary_)

            vocab = sorted(docs.fold_union(_part_vocab))
        self.vocabulary_ = {w: i for i, w in enumerate(vocab)}
        return self

    def transform(self, X):
        check_is_fitted(self, "vocabulary_")
        docs = _as_distlist(X)
        vec = _SkCount(lowercase=self.lowercase,
                       ngram_range=self.ngram_range, binary=self.binary,
                       vocabulary=self.vocabulary_)
        return docs.map_partition(
            lambda part: [vec.transform(part)] if part else [])

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)

    def get_feature_names_out(self):
        inv = sorted(self.vocabulary_, key=self.vocabulary_.get)
        return np.asarray(inv, dtype=object)
