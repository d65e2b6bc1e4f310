"""mi355ml — MI355X-native distributed classical ML.

A from-scratch framework with the capabilities of the dask-ml lineage
(sklearn-compatible .fit/.transform/.predict estimators, joblib persistence),
re-designed MI355X-first: row-sharded device arrays (one process per GPU),
RCCL collectives over xGMI instead of task-graph tree-reductions, and
hand-written CDNA4 HIP kernels for the hot per-shard math.

Reference capability map: SURVEY.md (mrocklin/dask-ml lineage).
"""

__version__ = "0.1.0"

from . import config
from .core.array import DistArray, concat, from_numpy, shard_rows, take
from .core.colarray import ColDistArray, from_numpy_cols, shard_cols
from .core.bag import DistList
from .parallel import comm

from . import (cluster, compose, dataframe, datasets, decomposition, diagnostics, ensemble,
               feature_extraction, impute, linear_model, metrics,
               model_selection, naive_bayes, preprocessing, utils, wrappers)

__all__ = [
    "config", "comm", "DistArray", "DistList", "from_numpy", "shard_rows",
    "ColDistArray", "from_numpy_cols", "shard_cols", "take", "concat",
    "cluster", "compose", "dataframe", "datasets", "decomposition", "diagnostics", "ensemble",
    "feature_extraction", "impute", "linear_model", "metrics",
    "model_selection", "naive_bayes", "preprocessing", "utils", "wrappers",
    "__version__",
]
