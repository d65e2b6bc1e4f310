"""Categorizer/DummyEncoder (dataframe partitions) + checkpoint/resume +
native GLM partial_fit."""

import numpy as np
import pandas as pd
import torch

from mi355ml.core.array import from_numpy
from mi355ml.core.bag import DistList
from mi355ml.preprocessing import Categorizer, DummyEncoder


def _frames():
    df = pd.DataFrame({
        "a": ["x", "y", "x", "z", "y", "x"],
        "b": [1.0, 2.0, 3.0, 4.0, 5.0, 6.0],
    })
    return DistList.from_global([df.iloc[:3], df.iloc[3:]])


def test_categorizer_dummy_encoder():
    frames = _frames()
    cz = Categorizer().fit(frames)
    assert cz.categories_ == {"a": ["x", "y", "z"]}
    cat = cz.transform(frames)
    for f in cat.gather():
        assert isinstance(f["a"].dtype, pd.CategoricalDtype)
        assert list(f["a"].cat.categories) == ["x", "y", "z"]
    de = DummyEncoder().fit(cat)
    enc = de.transform(cat)
    out = pd.concat(enc.gather())
    assert {"a_x", "a_y", "a_z", "b"} <= set(out.columns)
    assert out.shape == (6, 4)


def test_checkpoint_roundtrip(tmp_path):
    from mi355ml.checkpoint import save_checkpoint, load_checkpoint
    p = str(tmp_path / "s.ckpt")
    save_checkpoint(p, {"iteration": 5, "centers": torch.ones(3, 2)})
    st = load_checkpoint(p)
    assert st["iteration"] == 5
    np.testing.assert_array_equal(st["centers"].cpu().numpy(), np.ones((3, 2)))


def test_kmeans_checkpoint_resume(tmp_path):
    from mi355ml.cluster import KMeans
    rng = np.random.default_rng(0)
    centers = rng.uniform(-8, 8, (3, 4))
    lab = rng.integers(0, 3, 1200)
    X = (centers[lab] + rng.standard_normal((1200, 4))).astype(np.float32)
    p = str(tmp_path / "km.ckpt")
    km1 = KMeans(n_clusters=3, random_state=0, max_iter=3,
                 checkpoint_path=p, checkpoint_every=1)
    km1.fit(from_numpy(X))
    import os
    assert os.path.exists(p)
    km2 = KMeans(n_clusters=3, random_state=0, max_iter=50,
                 checkpoint_path=p, checkpoint_every=1)
    km2.fit(from_numpy(X))  # resumes from iteration 3
    assert km2.n_iter_ >= 3
    assert km2.inertia_ <= km1.inertia_ * 1.001


def test_native_glm_partial_fit_incremental():
    from mi355ml.linear_model import LogisticRegression
    rng = np.random.default_rng(1)
    X = rng.standard_normal((2000, 5)).astype(np.float32)
    w = rng.standard_normal(5) * 2
    y = ((1 / (1 + np.exp(-(X @ w)))) > rng.uniform(size=2000)).astype(
        np.float32)
    lr = LogisticRegression(C=1e3)
    for _ in range(30):
        lr.partial_fit(from_numpy(X), from_numpy(y))
    assert lr.score(from_numpy(X), from_numpy(y)) > 0.85


def test_dataframe_groupby_agg():
    from mi355ml import dataframe as mdf
    rng = np.random.default_rng(7)
    df = pd.DataFrame({
        "g": rng.choice(["a", "b", "c"], 1000),
        "v": rng.standard_normal(1000),
        "w": rng.integers(0, 10, 1000),
    })
    frames = mdf.from_pandas(df, npartitions=4)
    for agg in ("sum", "count", "mean", "min", "max"):
        got = mdf.groupby_agg(frames, "g", agg=agg, columns=["v", "w"])
        if agg == "count":
            ref = df.groupby("g")[["v", "w"]].count()
        elif agg == "sum":
            ref = df.groupby("g")[["v", "w"]].sum(numeric_only=True)
        else:
            ref = getattr(df.groupby("g")[["v", "w"]], agg)(
                numeric_only=(agg == "sum"))
        pd.testing.assert_frame_equal(got.sort_index(), ref.sort_index(),
                                      check_dtype=False, check_like=True)
    vc = mdf.value_counts(frames, "g")
    pd.testing.assert_series_equal(
        vc.sort_index(), df["g"].value_counts().sort_index(),
        check_names=False, check_dtype=False)
    full = mdf.concat(frames)
    assert len(full) == 1000


def _df_world2(rank, world):
    from mi355ml import dataframe as mdf
    df = pd.DataFrame({"g": ["x", "y"] * 50, "v": np.arange(100.0)})
    frames = mdf.from_pandas(df, npartitions=2)
    assert len(frames.local) == 1
    got = mdf.groupby_agg(frames, "g", agg="mean", columns=["v"])
    ref = df.groupby("g")[["v"]].mean()
    pd.testing.assert_frame_equal(got.sort_index(), ref.sort_index(),
                                  check_dtype=False)


def test_dataframe_world2(mp_runner):
    mp_runner(_df_world2, world=2)


def _toy_frames(n=500, seed=0, nparts=3):
    from mi355ml.dataframe import from_pandas
    rng = np.random.default_rng(seed)
    df = pd.DataFrame({
        "k": rng.integers(0, 17, n),
        "v": rng.standard_normal(n),
        "w": rng.integers(0, 5, n),
    })
    return df, from_pandas(df, npartitions=nparts)


def test_dataframe_shuffle_groupby_apply():
    from mi355ml.dataframe import concat, groupby_apply, shuffle
    df, frames = _toy_frames()
    sh = shuffle(frames, "k")
    got = concat(sh).sort_values(["k", "v"]).reset_index(drop=True)
    exp = df.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp)
    # groupby-apply: demean within each key group (needs co-located groups)
    out = groupby_apply(frames, "k", lambda g: g.assign(v=g.v - g.v.mean()))
    got = concat(out)
    exp2 = pd.concat([g.assign(v=g.v - g.v.mean())
                      for _, g in df.groupby("k", group_keys=False)])
    got = got.sort_values(["k", "v"]).reset_index(drop=True)
    exp2 = exp2.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp2)


def test_dataframe_merge():
    from mi355ml.dataframe import concat, from_pandas, merge
    df, frames = _toy_frames(300, seed=1)
    rng = np.random.default_rng(2)
    other = pd.DataFrame({"k": np.arange(17), "name": rng.integers(0, 9, 17)})
    oframes = from_pandas(other, npartitions=2)
    for how in ("inner", "left", "outer"):
        got = concat(merge(frames, oframes, on="k", how=how))
        exp = df.merge(other, on="k", how=how)
        got = got.sort_values(["k", "v"]).reset_index(drop=True)
        exp = exp.sort_values(["k", "v"]).reset_index(drop=True)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _shuffle_world2(rank, world, n):
    from mi355ml.dataframe import concat, groupby_agg, merge, shuffle
    from mi355ml.dataframe import from_pandas
    rng = np.random.default_rng(5)
    df = pd.DataFrame({"k": rng.integers(0, 11, n),
                       "v": rng.standard_normal(n)})
    frames = from_pandas(df, npartitions=4)
    sh = shuffle(frames, "k")
    # disjoint key ownership across ranks
    local_keys = set()
    for p in sh.local:
        local_keys |= set(p["k"].unique().tolist())
    others = sh.comm.all_gather_object(local_keys)
    assert local_keys.isdisjoint(others[1 - rank])
    got = concat(sh).sort_values(["k", "v"]).reset_index(drop=True)
    exp = df.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp)


def test_dataframe_shuffle_world2(mp_runner):
    mp_runner(_shuffle_world2, world=2, args=(400,))


def _merge_world2(rank, world, n):
    from mi355ml.dataframe import concat, from_pandas, merge
    rng = np.random.default_rng(6)
    left = pd.DataFrame({"k": rng.integers(0, 9, n),
                         "v": rng.standard_normal(n)})
    right = pd.DataFrame({"k": np.arange(9), "w": np.arange(9) * 10})
    got = concat(merge(from_pandas(left, npartitions=3),
                       from_pandas(right, npartitions=2), on="k"))
    exp = left.merge(right, on="k")
    got = got.sort_values(["k", "v"]).reset_index(drop=True)
    exp = exp.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dataframe_merge_world2(mp_runner):
    mp_runner(_merge_world2, world=2, args=(300,))
