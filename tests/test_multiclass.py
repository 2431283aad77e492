"""
DistOneVsRestClassifier / DistOneVsOneClassifier tests (reference mirror:
skdist/distribute/tests/test_multiclass.py + SURVEY.md §4).
"""

import pickle

import numpy as np
import pytest
from sklearn.datasets import load_digits, load_iris
from sklearn.linear_model import LogisticRegression as SkLogReg
from sklearn.metrics import accuracy_score, f1_score

from skdist_amd import Cluster
from skdist_amd.distribute.multiclass import (
    DistOneVsOneClassifier,
    DistOneVsRestClassifier,
    _ConstantPredictor,
    _negatives_mask,
)
from skdist_amd.models import LogisticRegression


@pytest.fixture
def small_xy():
    X = np.array(
        [[1.0, 1.0, 1.0, 0.0], [0.0, 0.0, 0.0, 1.0],
         [-1.0, -1.0, -1.0, 0.5]] * 100
    )
    y = np.array([0, 1, 2] * 100)
    return X, y


def test_ovr_local(small_xy):
    X, y = small_xy
    clf = DistOneVsRestClassifier(SkLogReg(solver="liblinear"))
    clf.fit(X, y)
    assert np.array_equal(clf.predict(X[:3]), [0, 1, 2])
    p = clf.predict_proba(X[:3])
    assert p.shape == (3, 3)
    clf2 = pickle.loads(pickle.dumps(clf))
    assert np.array_equal(clf2.predict(X[:3]), [0, 1, 2])


def test_ovr_norm(small_xy):
    X, y = small_xy
    clf = DistOneVsRestClassifier(SkLogReg(solver="liblinear"), norm="l1")
    clf.fit(X, y)
    p = clf.predict_proba(X[:5])
    assert np.allclose(p.sum(axis=1), 1.0)


def test_ovo_local(small_xy):
    X, y = small_xy
    clf = DistOneVsOneClassifier(SkLogReg(solver="liblinear"))
    clf.fit(X, y)
    assert np.array_equal(clf.predict(X[:3]), [0, 1, 2])
    assert len(clf.estimators_) == 3  # 3 classes -> 3 pairs


def test_ovr_digits_quality():
    X, y = load_digits(return_X_y=True)
    clf = DistOneVsRestClassifier(SkLogReg(solver="liblinear"))
    clf.fit(X, y)
    f1 = f1_score(y, clf.predict(X), average="weighted")
    assert f1 > 0.95, f1  # reference: 0.9589 on held-out


def test_ovr_batched_cpu_cluster():
    X, y = load_iris(return_X_y=True)
    clf = DistOneVsRestClassifier(
        LogisticRegression(epochs=20, random_state=0), sc=Cluster()
    )
    clf.fit(X, y)
    acc = accuracy_score(y, clf.predict(X))
    assert acc > 0.9, acc
    assert len(clf.estimators_) == 3
    clf2 = pickle.loads(pickle.dumps(clf))
    assert accuracy_score(y, clf2.predict(X)) == acc


def test_ovo_batched_cpu_cluster():
    X, y = load_iris(return_X_y=True)
    clf = DistOneVsOneClassifier(
        LogisticRegression(epochs=20, random_state=0), sc=Cluster()
    )
    clf.fit(X, y)
    acc = accuracy_score(y, clf.predict(X))
    assert acc > 0.9, acc
    assert len(clf.estimators_) == 3


def test_ovo_batched_matches_generic():
    X, y = load_digits(return_X_y=True)
    est = LogisticRegression(epochs=15, random_state=0)
    a = DistOneVsOneClassifier(est, sc=Cluster()).fit(X, y)
    b = DistOneVsOneClassifier(est).fit(X, y)
    pa, pb = a.predict(X), b.predict(X)
    agree = (pa == pb).mean()
    assert agree > 0.97, agree  # same solver, device vs local path


def test_negatives_mask():
    y = np.array([1] * 10 + [0] * 90)
    m = _negatives_mask(y, 20, method="ratio", random_state=0)
    assert m[:10].all() and m.sum() == 30
    m = _negatives_mask(y, 2.0, method="multiplier", random_state=0)
    assert m.sum() == 30  # 10 pos + 2*10 neg
    m = _negatives_mask(y, 0.5, method="ratio", random_state=0)
    assert m.sum() == 10 + 45


def test_ovr_max_negatives(small_xy):
    X, y = small_xy
    clf = DistOneVsRestClassifier(
        SkLogReg(solver="liblinear"), max_negatives=50, random_state=3
    )
    clf.fit(X, y)
    assert np.array_equal(clf.predict(X[:3]), [0, 1, 2])


def test_constant_predictor():
    cp = _ConstantPredictor().fit(None, np.array([1]))
    assert np.array_equal(cp.predict(np.zeros((4, 2))), [1, 1, 1, 1])


def test_ovr_multilabel_sequences():
    X = np.array([[1.0, 0.0], [0.0, 1.0], [1.0, 1.0]] * 50)
    y = [("a",), ("b",), ("a", "b")] * 50
    clf = DistOneVsRestClassifier(SkLogReg(solver="liblinear"))
    clf.fit(X, y)
    preds = clf.predict(X[:3])
    assert set(preds[0]) == {"a"}
    assert set(preds[2]) == {"a", "b"}


def test_ovr_binary_y():
    """Binary y through OvR: single LabelBinarizer column must threshold,
    not argmax (regression: predictions collapsed to the first class);
    matches sklearn's OneVsRestClassifier exactly."""
    from sklearn.linear_model import LogisticRegression as SkLR
    from sklearn.multiclass import OneVsRestClassifier

    rng = np.random.default_rng(0)
    X = rng.standard_normal((200, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(int)
    ours = DistOneVsRestClassifier(SkLR(solver="liblinear")).fit(X, y)
    ref = OneVsRestClassifier(SkLR(solver="liblinear")).fit(X, y)
    assert (ours.predict(X) == ref.predict(X)).all()
    assert ours.predict_proba(X).shape == (200, 2)
    assert np.allclose(ours.predict_proba(X).sum(axis=1), 1.0)
    assert ours.decision_function(X).shape == (200,)
    # native estimator too
    nat = DistOneVsRestClassifier(
        LogisticRegression(epochs=10, random_state=0)
    ).fit(X, y)
    assert (nat.predict(X) == y).mean() > 0.9


def test_negatives_mask_modes():
    """ratio / multiplier / absolute cap semantics
    (reference multiclass.py:76-106)."""
    from skdist_amd.distribute.multiclass import _negatives_mask

    rng = np.random.default_rng(0)
    y = (rng.random(2000) < 0.05).astype(int)
    n_pos = int(y.sum())
    n_neg = 2000 - n_pos
    m = _negatives_mask(y, 0.5, method="ratio", random_state=0)
    assert m.sum() == n_pos + int(0.5 * n_neg)
    m = _negatives_mask(y, 3.0, method="multiplier", random_state=0)
    assert m.sum() == n_pos + 3 * n_pos
    m = _negatives_mask(y, 100, random_state=0)
    assert m.sum() == n_pos + 100
    # all positives always kept
    assert m[y == 1].all()


def test_ovo_decision_function_matches_sklearn():
    """OvO decision_function == sklearn's _ovr_decision_function
    (votes + bounded confidence tie-break), not raw confidence sums."""
    from sklearn.multiclass import OneVsOneClassifier

    rng = np.random.default_rng(0)
    X = rng.standard_normal((400, 8)).astype(np.float32)
    y = np.array(["aa", "bb", "cc", "dd"])[X[:, :4].argmax(axis=1)]
    ours = DistOneVsOneClassifier(SkLogReg(solver="liblinear")).fit(X, y)
    ref = OneVsOneClassifier(SkLogReg(solver="liblinear")).fit(X, y)
    np.testing.assert_allclose(
        ours.decision_function(X), ref.decision_function(X), atol=1e-12
    )
    assert (ours.predict(X) == ref.predict(X)).all()


def test_ovr_decision_function_matches_sklearn():
    """OvR decision_function = per-class RAW estimator scores (sklearn
    semantics), while predict_proba stays the normalized-proba matrix."""
    from sklearn.multiclass import OneVsRestClassifier

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 8)).astype(np.float32)
    y = X[:, :3].argmax(axis=1)
    ours = DistOneVsRestClassifier(SkLogReg(solver="liblinear")).fit(X, y)
    ref = OneVsRestClassifier(SkLogReg(solver="liblinear")).fit(X, y)
    np.testing.assert_allclose(
        ours.decision_function(X), ref.decision_function(X), atol=1e-12
    )
    yb = (X[:, 0] > 0).astype(int)
    ob = DistOneVsRestClassifier(SkLogReg(solver="liblinear")).fit(X, yb)
    rb = OneVsRestClassifier(SkLogReg(solver="liblinear")).fit(X, yb)
    assert ob.decision_function(X).shape == (300,)
    np.testing.assert_allclose(
        ob.decision_function(X), rb.decision_function(X), atol=1e-12
    )


def test_multilabel_thresholds_match_sklearn():
    """Multilabel indicator predictions: probability columns threshold
    at 0.5, raw decision columns at 0 (regression: a flat 0.5 threshold
    under-predicted positives for no-proba estimators)."""
    from sklearn.multiclass import OneVsRestClassifier
    from sklearn.svm import LinearSVC as SkSVC

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 8)).astype(np.float32)
    Y = np.zeros((300, 3), dtype=int)
    Y[:, 0] = X[:, 0] > 0
    Y[:, 1] = X[:, 1] > 0.5
    Y[:, 2] = (X[:, 0] + X[:, 1]) > 0
    for est_f in (SkSVC, lambda: SkLogReg(solver="liblinear")):
        ours = DistOneVsRestClassifier(est_f()).fit(X, Y)
        ref = OneVsRestClassifier(est_f()).fit(X, Y)
        np.testing.assert_array_equal(
            np.asarray(ours.predict(X)), ref.predict(X)
        )
