"""Text pipeline: HashingVectorizer -> Incremental(SGD) over sparse
partitions, then serve the model over HTTP."""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.join(_os.path.dirname(
    _os.path.abspath(__file__)), ".."))
import numpy as np
from sklearn.linear_model import SGDClassifier

from mi355ml.core.bag import DistList
from mi355ml.feature_extraction.text import HashingVectorizer
from mi355ml.wrappers import Incremental

docs = ["good great fine", "bad awful poor", "nice excellent happy",
        "terrible sad worst"] * 1000
labels = np.array([1, 0, 1, 0] * 1000)
Xs = HashingVectorizer(n_features=2 ** 18).transform(DistList.from_global(docs))
ys = DistList.from_global([labels])
clf = Incremental(SGDClassifier(loss="log_loss", random_state=0),
                  shuffle_blocks=False)
clf.fit(Xs, ys, epochs=3)
print("train acc:",
      np.mean(np.concatenate([np.asarray(p) for p in
                              clf.predict(Xs).gather()]) == labels))
# joblib.dump(clf, "model.joblib"); python -m mi355ml.serving model.joblib
