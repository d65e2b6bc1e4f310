"""End-to-end workflow demo at scale (1 GPU): generate 5e7x100 -> scale ->
PCA(20) -> LogisticRegression -> persist -> reload -> serve-style predict.
"""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import joblib
import numpy as np
import torch

from mi355ml import datasets
from mi355ml.decomposition import PCA
from mi355ml.linear_model import LogisticRegression
from mi355ml.model_selection import train_test_split
from mi355ml.preprocessing import StandardScaler
from sklearn.pipeline import Pipeline


def t(name, fn):
    t0 = time.perf_counter()
    out = fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    print(f"{name:44s} {time.perf_counter()-t0:7.2f} s", flush=True)
    return out


n = 50_000_000 if torch.cuda.is_available() else 100_000
X, y = t(f"generate {n:.0e}x100", lambda: datasets.make_classification(
    n_samples=n, n_features=100, n_informative=30, random_state=0))
Xtr, Xte, ytr, yte = t("train_test_split 80/20", lambda: train_test_split(
    X, y, test_size=0.2, random_state=0))
pipe = Pipeline([
    ("scale", StandardScaler()),
    ("pca", PCA(n_components=20, svd_solver="randomized")),
    ("clf", LogisticRegression(C=1.0, max_iter=100)),
])
t("pipeline fit (scale->PCA20->logreg)", lambda: pipe.fit(Xtr, ytr))
acc = t("pipeline score on held-out 20%", lambda: pipe.score(Xte, yte))
print(f"    held-out accuracy: {acc:.4f}")
t("joblib dump+load roundtrip", lambda: joblib.dump(pipe, "/tmp/pipe.joblib"))
pipe2 = joblib.load("/tmp/pipe.joblib")
sample = Xte.local[:16].cpu().numpy()
pred = pipe2.predict(sample)
print(f"    reloaded model predicts {int(np.asarray(pred.gather().cpu()).sum())}"
      f"/16 positives on a sample batch")
