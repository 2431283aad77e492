"""
Ensemble tests (reference mirror: skdist/distribute/tests/test_ensemble.py
+ test_spark.py:50-76; BASELINE.md quality rows for breast_cancer).
"""

import pickle

import numpy as np
import pytest
from sklearn.datasets import load_breast_cancer
from sklearn.metrics import f1_score, r2_score, roc_auc_score

from skdist_amd import Cluster
from skdist_amd.distribute.ensemble import (
    DistExtraTreesClassifier,
    DistExtraTreesRegressor,
    DistRandomForestClassifier,
    DistRandomForestRegressor,
    DistRandomTreesEmbedding,
)


@pytest.fixture
def small_clf():
    X = np.array([[1.0, 1.0, 1.0], [0.0, 0.0, 0.0], [-1.0, -1.0, -1.0]] * 100)
    y = np.array([0, 0, 1] * 100)
    return X, y


@pytest.fixture
def small_reg():
    rng = np.random.default_rng(5)
    X = rng.normal(size=(300, 4))
    y = X[:, 0] * 2 - X[:, 1] + 0.1 * rng.normal(size=300)
    return X, y


def test_rf_classifier(small_clf):
    X, y = small_clf
    clf = DistRandomForestClassifier(n_estimators=10, random_state=3)
    clf.fit(X, y)
    assert np.allclose(clf.predict(X[:3]), [0, 0, 1])
    p = clf.predict_proba(X[:3])
    assert p.shape == (3, 2)
    assert np.allclose(p.sum(axis=1), 1.0)
    clf2 = pickle.loads(pickle.dumps(clf))
    assert np.allclose(clf2.predict(X[:3]), [0, 0, 1])


def test_extra_trees_classifier(small_clf):
    X, y = small_clf
    clf = DistExtraTreesClassifier(n_estimators=10, random_state=3)
    clf.fit(X, y)
    assert np.allclose(clf.predict(X[:3]), [0, 0, 1])


def test_rf_regressor(small_reg):
    X, y = small_reg
    reg = DistRandomForestRegressor(n_estimators=20, random_state=3)
    reg.fit(X, y)
    assert r2_score(y, reg.predict(X)) > 0.8


def test_extra_trees_regressor(small_reg):
    X, y = small_reg
    reg = DistExtraTreesRegressor(n_estimators=20, random_state=3)
    reg.fit(X, y)
    assert r2_score(y, reg.predict(X)) > 0.8


def test_embedding_shape():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(30, 3))
    emb = DistRandomTreesEmbedding(n_estimators=10, random_state=5)
    T = emb.fit_transform(X)
    assert T.shape[0] == 30
    assert T.shape[1] == emb.apply(X).max() * 0 + T.shape[1]  # consistent
    T2 = emb.transform(X)
    assert (T != T2).nnz == 0


def test_rf_quality_breast_cancer():
    """Reference quality row: RF roc_auc 0.997 / f1w 0.986 on train."""
    X, y = load_breast_cancer(return_X_y=True)
    clf = DistRandomForestClassifier(n_estimators=50, random_state=0)
    clf.fit(X, y)
    assert roc_auc_score(y, clf.predict_proba(X)[:, 1]) > 0.995
    assert f1_score(y, clf.predict(X)) > 0.98


def test_oob_score(small_clf):
    X, y = small_clf
    clf = DistRandomForestClassifier(
        n_estimators=20, oob_score=True, random_state=0
    )
    clf.fit(X, y)
    assert clf.oob_score_ > 0.9
    assert clf.oob_decision_function_.shape == (len(y), 2)


def test_warm_start(small_clf):
    X, y = small_clf
    clf = DistRandomForestClassifier(
        n_estimators=5, warm_start=True, random_state=0
    )
    clf.fit(X, y)
    clf.n_estimators = 9
    clf.fit(X, y)
    assert len(clf.estimators_) == 9


def test_forest_cluster(small_clf):
    X, y = small_clf
    clf = DistRandomForestClassifier(
        n_estimators=8, random_state=3, sc=Cluster()
    )
    clf.fit(X, y)
    assert np.allclose(clf.predict(X[:3]), [0, 0, 1])
    assert clf.sc is None


def test_feature_importances(small_clf):
    X, y = small_clf
    clf = DistRandomForestClassifier(n_estimators=10, random_state=0)
    clf.fit(X, y)
    fi = clf.feature_importances_
    assert fi.shape == (3,)
    assert np.isclose(fi.sum(), 1.0)


def test_device_gate_fallbacks():
    """Unsupported params must route to the CPU per-tree path (sklearn
    trees), silently and correctly, even when a Cluster is given."""
    from skdist_amd import Cluster
    from skdist_amd.distribute.ensemble import DistRandomForestClassifier

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)
    sc = Cluster()  # CPU device here -> _device_fit_ok False either way

    for kwargs in (
        {"max_leaf_nodes": 8},
        {"class_weight": "balanced"},
        {"criterion": "log_loss"},
    ):
        clf = DistRandomForestClassifier(
            sc=sc, n_estimators=4, random_state=0, **kwargs)
        clf.fit(X, y)
        assert len(clf.estimators_) == 4
        assert clf.score(X, y) > 0.8
    # sample_weight also falls back
    clf = DistRandomForestClassifier(sc=sc, n_estimators=4, random_state=0)
    clf.fit(X, y, sample_weight=np.ones(len(y)))
    assert len(clf.estimators_) == 4


def test_warm_start_adds_trees():
    from skdist_amd.distribute.ensemble import DistRandomForestClassifier

    rng = np.random.default_rng(1)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)
    clf = DistRandomForestClassifier(
        n_estimators=4, warm_start=True, random_state=0)
    clf.fit(X, y)
    assert len(clf.estimators_) == 4
    clf.n_estimators = 7
    clf.fit(X, y)
    assert len(clf.estimators_) == 7


def test_warm_start_new_trees_get_fresh_seeds():
    """Warm-started additions must not duplicate existing trees
    (regression: the redrawn seed stream was sliced from the front)."""
    rng = np.random.default_rng(0)
    X = rng.standard_normal((200, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(int)
    rf = DistRandomForestClassifier(
        n_estimators=3, random_state=0, warm_start=True
    ).fit(X, y)
    first = list(rf._seeds)
    rf.set_params(n_estimators=6)
    rf.fit(X, y)
    second = list(rf._seeds)
    assert len(rf.estimators_) == 6
    assert not (set(first) & set(second))


def test_get_oof_helpers():
    """Out-of-fold helpers (reference ensemble.py:112-151): oof rows come
    from fits that never saw them; the returned model is fully fitted."""
    from skdist_amd.distribute.ensemble import get_oof, get_single_oof

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(int)
    clf = DistRandomForestClassifier(n_estimators=10, random_state=0)
    fitted, oof = get_oof(clf, X, y, n_splits=3)
    assert oof.shape == (300, 2)
    assert np.allclose(oof.sum(axis=1), 1.0)
    assert len(fitted.estimators_) == 10
    # oof probabilities are predictive (out-of-sample accuracy)
    assert ((oof[:, 1] > 0.5).astype(int) == y).mean() > 0.8
    idx_tr = np.arange(150)
    idx_te = np.arange(150, 300)
    te, proba = get_single_oof(clf, X, y, idx_tr, idx_te)
    assert (te == idx_te).all() and proba.shape == (150, 2)
