from ._split import KFold, ShuffleSplit, check_cv, train_test_split
from ._search import GridSearchCV, RandomizedSearchCV
from ._incremental import IncrementalSearchCV
from ._successive_halving import SuccessiveHalvingSearchCV
from ._hyperband import HyperbandSearchCV

__all__ = [
    "train_test_split", "KFold", "ShuffleSplit", "check_cv",
    "GridSearchCV", "RandomizedSearchCV",
    "IncrementalSearchCV", "SuccessiveHalvingSearchCV", "HyperbandSearchCV",
]
