"""Hyperband over SGDClassifier.partial_fit (BASELINE config 5):
trials fan out over the ranks, data stays sharded in place."""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.join(_os.path.dirname(
    _os.path.abspath(__file__)), ".."))
from sklearn.linear_model import SGDClassifier

from mi355ml import datasets
from mi355ml.model_selection import HyperbandSearchCV
from mi355ml.parallel import comm

c = comm.init()
X, y = datasets.make_classification(n_samples=1_000_000, n_features=50,
                                    n_informative=10, random_state=0)
search = HyperbandSearchCV(
    SGDClassifier(loss="log_loss", random_state=0),
    {"alpha": [10 ** -i for i in range(2, 7)],
     "eta0": [0.01, 0.1, 1.0],
     "learning_rate": ["constant", "invscaling"]},
    max_iter=81, random_state=0)
search.fit(X, y)
if c.rank == 0:
    print("best:", search.best_params_, f"score={search.best_score_:.4f}",
          "models:", search.metadata_["n_models"])
comm.shutdown()
