"""Histogram forest builder tests.

CPU tier exercises the eager (torch) engine — the numerics reference the
HIP kernels are asserted against in the gpu tier (test_forest_gpu.py).
Reference behavior being mirrored: skdist/distribute/ensemble.py (one
tree per task, bootstrap weights via randint+bincount, forest aggregation
by proba mean).
"""

import numpy as np
import pickle

import pytest
from sklearn.ensemble import RandomForestClassifier, RandomForestRegressor

from skdist_amd.models.forest import (
    BinnedDataset,
    ForestBuilder,
    HistTree,
    resolve_max_features,
)


def _cls_data(n=3000, f=16, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f)
    y = ((X @ w + np.sin(3 * X[:, 0]) * 2
          + 0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
    return X, y


def test_resolve_max_features():
    assert resolve_max_features(None, 64) == 64
    assert resolve_max_features("sqrt", 64) == 8
    assert resolve_max_features("log2", 64) == 6
    assert resolve_max_features(0.5, 64) == 32
    assert resolve_max_features(10, 64) == 10
    assert resolve_max_features(100, 64) == 64


def test_binning_roundtrip():
    X, y = _cls_data(500, 4)
    ds = BinnedDataset(X, y, "cpu", is_cls=True, nbins=16)
    # code <= b  ⟺  x <= edges[b]
    edges = ds.edges_np()
    codes = ds.codes.numpy()
    for j in range(4):
        for b in (0, 7, 14):
            lhs = codes[:, j] <= b
            rhs = X[:, j] <= edges[j, b]
            assert (lhs == rhs).all()


def test_classifier_quality_vs_sklearn():
    X, y = _cls_data()
    Xtr, Xte, ytr, yte = X[:2200], X[2200:], y[:2200], y[2200:]
    ds = BinnedDataset(Xtr, ytr, "cpu", is_cls=True)
    b = ForestBuilder(ds, "gini", max_features="sqrt", bootstrap=True,
                      tree_batch=16)
    trees = b.build(list(range(40)))
    proba = np.mean([t.predict_proba(Xte) for t in trees], axis=0)
    acc = (ds.classes_[proba.argmax(1)] == yte).mean()
    ref = RandomForestClassifier(n_estimators=40, random_state=0)
    ref_acc = ref.fit(Xtr, ytr).score(Xte, yte)
    assert acc > ref_acc - 0.05, (acc, ref_acc)


def test_regressor_quality_vs_sklearn():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((2500, 10)).astype(np.float32)
    w = rng.standard_normal(10)
    y = (X @ w + 0.1 * rng.standard_normal(2500)).astype(np.float32)
    Xtr, Xte, ytr, yte = X[:2000], X[2000:], y[:2000], y[2000:]
    ds = BinnedDataset(Xtr, ytr, "cpu", is_cls=False)
    b = ForestBuilder(ds, "squared_error", max_features=1.0,
                      bootstrap=True, tree_batch=16)
    trees = b.build(list(range(30)))
    pred = np.mean([t.predict(Xte) for t in trees], axis=0)
    r2 = 1 - ((pred - yte) ** 2).sum() / ((yte - yte.mean()) ** 2).sum()
    ref = RandomForestRegressor(n_estimators=30, random_state=0)
    ref_r2 = ref.fit(Xtr, ytr).score(Xte, yte)
    assert r2 > ref_r2 - 0.1, (r2, ref_r2)


def test_extra_mode_and_depth_cap():
    X, y = _cls_data(1200, 8)
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    b = ForestBuilder(ds, "gini", max_depth=3, max_features="sqrt",
                      bootstrap=False, extra_mode=True, tree_batch=8)
    trees = b.build(list(range(8)))
    for t in trees:
        # depth <= 3  ⇒  <= 2^4 - 1 nodes
        assert t.node_count <= 15
    proba = np.mean([t.predict_proba(X) for t in trees], axis=0)
    assert (ds.classes_[proba.argmax(1)] == y).mean() > 0.7


def test_min_samples_and_impurity_controls():
    X, y = _cls_data(800, 6)
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    big = ForestBuilder(ds, "gini", bootstrap=False).build([7])[0]
    small = ForestBuilder(ds, "gini", bootstrap=False,
                          min_samples_leaf=100).build([7])[0]
    assert small.node_count < big.node_count
    tiny = ForestBuilder(ds, "gini", bootstrap=False,
                         min_impurity_decrease=0.05).build([7])[0]
    assert tiny.node_count < big.node_count


def test_entropy_criterion():
    X, y = _cls_data(1000, 8)
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    trees = ForestBuilder(ds, "entropy", bootstrap=True,
                          tree_batch=4).build(list(range(8)))
    proba = np.mean([t.predict_proba(X) for t in trees], axis=0)
    assert (ds.classes_[proba.argmax(1)] == y).mean() > 0.85


def test_multiclass():
    rng = np.random.default_rng(3)
    X = rng.standard_normal((1500, 8)).astype(np.float32)
    y = (X[:, 0] * 2 + X[:, 1]).astype(np.int64) % 4
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    assert ds.S == 4
    trees = ForestBuilder(ds, "gini", bootstrap=True,
                          tree_batch=8).build(list(range(16)))
    proba = np.mean([t.predict_proba(X) for t in trees], axis=0)
    assert proba.shape == (1500, 4)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, atol=1e-5)


def test_hist_tree_pickles_and_applies():
    X, y = _cls_data(600, 5)
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    t = ForestBuilder(ds, "gini", bootstrap=True).build([11])[0]
    t2 = pickle.loads(pickle.dumps(t))
    np.testing.assert_array_equal(t.predict(X), t2.predict(X))
    leaves = t.apply(X)
    assert (t.feature[leaves] == -1).all()
    assert np.isclose(t.feature_importances_.sum(), 1.0)


def test_bootstrap_weights_deterministic():
    X, y = _cls_data(400, 4)
    ds = BinnedDataset(X, y, "cpu", is_cls=True)
    b = ForestBuilder(ds, "gini", bootstrap=True)
    w1 = b.make_weights([5, 9])
    w2 = b.make_weights([5, 9])
    assert (w1 == w2).all()
    assert (w1[0] != w1[1]).any()
    # multinomial: total draws == n
    assert int(w1[0].to(int).sum()) == 400


def test_sklearn_tree_flattening_host():
    """sklearn fitted trees flatten onto the HistTree layout with exact
    prediction parity (host traversal; the GPU kernel walks the same
    arrays — tested in test_forest_gpu)."""
    from sklearn.tree import DecisionTreeClassifier, DecisionTreeRegressor

    from skdist_amd.models.forest import _sklearn_tree_to_hist_tree

    X, y = _cls_data(800, 6)
    dt = DecisionTreeClassifier(max_depth=6, random_state=0).fit(X, y)
    ht = _sklearn_tree_to_hist_tree(dt, "proba")
    np.testing.assert_allclose(
        ht.predict_proba(X), dt.predict_proba(X), atol=1e-6)

    yr = (X[:, 0] * 2 + X[:, 1]).astype(np.float32)
    dr = DecisionTreeRegressor(max_depth=6, random_state=0).fit(X, yr)
    hr = _sklearn_tree_to_hist_tree(dr, "value")
    np.testing.assert_allclose(hr.predict(X), dr.predict(X), atol=1e-5)
