"""Distributed LogisticRegression on synthetic data.

1 GPU:   python examples/logistic_regression.py
8 GPUs:  python -m mi355ml.parallel.launch --gpus 8 examples/logistic_regression.py
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.join(_os.path.dirname(
    _os.path.abspath(__file__)), ".."))
from mi355ml import datasets
from mi355ml.linear_model import LogisticRegression
from mi355ml.model_selection import train_test_split
from mi355ml.parallel import comm

c = comm.init()
X, y = datasets.make_classification(n_samples=10_000_000, n_features=100,
                                    n_informative=20, random_state=0)
Xtr, Xte, ytr, yte = train_test_split(X, y, test_size=0.2, random_state=0)
clf = LogisticRegression(C=1.0, solver="lbfgs", max_iter=100).fit(Xtr, ytr)
if c.rank == 0:
    print(f"iters={clf.n_iter_} test-acc={clf.score(Xte, yte):.4f}")
comm.shutdown()
