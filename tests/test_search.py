"""
CPU-path tests for DistGridSearchCV / DistRandomizedSearchCV — mirrors the
reference test strategy (skdist/distribute/tests/test_search.py and
SURVEY.md §4): behavioral asserts on tiny data through the sc=None branch,
plus pickle round-trips and cv_results_ schema checks.
"""

import pickle

import numpy as np
import pytest
from scipy.stats import uniform
from sklearn.datasets import load_iris
from sklearn.linear_model import LogisticRegression as SkLogReg

from skdist_amd.distribute.search import (
    DistGridSearchCV,
    DistRandomizedSearchCV,
)


@pytest.fixture
def small_xy():
    X = np.array([[1.0, 1.0, 1.0], [0.0, 0.0, 0.0], [-1.0, -1.0, -1.0]] * 100)
    y = np.array([0, 0, 1] * 100)
    return X, y


def test_grid_search_local(small_xy):
    X, y = small_xy
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [0.1, 1.0, 10.0]}, cv=5
    )
    gs.fit(X, y)
    preds = gs.predict(X[:3])
    assert np.allclose(preds, [0, 0, 1])
    assert gs.best_score_ > 0.9


def test_randomized_search_local(small_xy):
    X, y = small_xy
    rs = DistRandomizedSearchCV(
        SkLogReg(solver="liblinear"), {"C": uniform(0.1, 10)},
        cv=5, n_iter=4, random_state=0,
    )
    rs.fit(X, y)
    preds = rs.predict(X[:3])
    assert np.allclose(preds, [0, 0, 1])


def test_cv_results_schema(small_xy):
    X, y = small_xy
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [0.1, 1.0]}, cv=3
    )
    gs.fit(X, y)
    r = gs.cv_results_
    for key in (
        "mean_fit_time", "std_fit_time", "mean_score_time", "std_score_time",
        "params", "param_C", "mean_test_score", "std_test_score",
        "rank_test_score", "split0_test_score", "split2_test_score",
    ):
        assert key in r, key
    assert len(r["params"]) == 2
    assert r["rank_test_score"].dtype == np.int32
    # ranks: best candidate has rank 1
    assert r["rank_test_score"].min() == 1


def test_iris_baseline_config():
    """BASELINE.json config 1: LogReg 4-point C grid, 3-fold, sc=None."""
    X, y = load_iris(return_X_y=True)
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"),
        {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3,
    )
    gs.fit(X, y)
    assert gs.best_score_ > 0.9
    assert len(gs.cv_results_["params"]) == 4


def test_pickle_contract(small_xy):
    """Fitted search strips sc and pickles (reference search.py:568-570)."""
    X, y = small_xy
    gs = DistGridSearchCV(SkLogReg(solver="liblinear"), {"C": [1.0]}, cv=3)
    gs.fit(X, y)
    assert gs.sc is None
    blob = pickle.dumps(gs)
    gs2 = pickle.loads(blob)
    assert np.allclose(gs2.predict(X[:3]), [0, 0, 1])


def test_multimetric(small_xy):
    X, y = small_xy
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [0.1, 1.0]},
        scoring=["accuracy", "roc_auc"], refit="roc_auc", cv=3,
    )
    gs.fit(X, y)
    assert "mean_test_roc_auc" in gs.cv_results_
    assert "mean_test_accuracy" in gs.cv_results_
    assert gs.best_score_ > 0.9


def test_error_score(small_xy):
    X, y = small_xy

    class Broken(SkLogReg):
        def fit(self, *a, **k):
            raise RuntimeError("boom")

    gs = DistGridSearchCV(
        Broken(), {"C": [1.0]}, cv=3, error_score=0.0, refit=False
    )
    with pytest.warns(Warning):
        gs.fit(X, y)
    assert np.allclose(gs.cv_results_["mean_test_score"], 0.0)

    gs = DistGridSearchCV(Broken(), {"C": [1.0]}, cv=3, error_score="raise")
    with pytest.raises(RuntimeError):
        gs.fit(X, y)


def test_n_jobs_parallel(small_xy):
    X, y = small_xy
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [0.1, 1.0]}, cv=3, n_jobs=2
    )
    gs.fit(X, y)
    assert gs.best_score_ > 0.9


def test_preds_attribute(small_xy):
    X, y = small_xy
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [1.0]}, cv=3, preds=True
    )
    gs.fit(X, y)
    assert gs.preds_.shape[0] == len(y)


def test_non_partition_cv_falls_back(small_xy):
    """ShuffleSplit folds don't partition the rows -> the batched device
    solve must fall back to the generic per-task path (same results
    contract)."""
    from sklearn.model_selection import ShuffleSplit

    from skdist_amd import Cluster
    from skdist_amd.models import LogisticRegression

    X, y = small_xy
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]},
        cv=ShuffleSplit(n_splits=3, test_size=0.3, random_state=0),
        sc=Cluster(),
    )
    gs.fit(X, y)
    assert gs.best_score_ > 0.7
    assert len(gs.cv_results_["mean_test_score"]) == 2


def test_out_of_fold_preds(small_xy):
    """preds=True returns out-of-fold predictions row-aligned with X
    (reference search.py:551-560)."""
    from skdist_amd.models import LogisticRegression

    X, y = small_xy
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3, preds=True, sc=None)
    gs.fit(X, y)
    assert gs.preds_.shape[0] == len(y)
    # proba columns are row-aligned: argmax should mostly match labels
    agree = (gs.classes_[gs.preds_.argmax(axis=1)] == y).mean()
    assert agree > 0.8


def test_search_with_sample_weight_batched(small_xy):
    """fit_params={'sample_weight': w} through the search.  As of
    round 2 an explicit sample_weight routes to the GENERIC path
    (sklearn forwards it to the scorers too, test-sliced — the device
    scorers are unweighted); per-task fits still use the fused weight
    plane via est.fit(sample_weight=...)."""
    from skdist_amd import Cluster
    from skdist_amd.models import LogisticRegression

    X, y = small_xy
    w = np.ones(len(y))
    w[::4] = 3.0
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3, sc=Cluster())
    gs.fit(X, y, sample_weight=w)
    assert gs.best_score_ > 0.7
    # zero-weight garbage rows don't change results
    w0 = np.ones(len(y)); w0[::3] = 0.0
    yb = y.copy(); yb[::3] = 1 - yb[::3] if set(y) == {0, 1} else yb[::3]
    g1 = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [1.0]}, cv=3, sc=Cluster())
    g1.fit(X, y, sample_weight=w0)
    g2 = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [1.0]}, cv=3, sc=Cluster())
    g2.fit(X, yb, sample_weight=w0)
    # the fitted models must be identical: zero-weight rows have no
    # gradient (and with weighted scoring they don't count there either)
    np.testing.assert_allclose(
        g1.best_estimator_.coef_, g2.best_estimator_.coef_, atol=1e-7)


def test_search_sample_weight_matches_sklearn_exactly():
    """Generic-path fit AND scoring with sample_weight reproduce
    sklearn's GridSearchCV bit-for-bit (round-2 fix: fold-sliced fit
    params + weighted scorer forwarding; previously a shape crash)."""
    from sklearn.linear_model import LogisticRegression as SkLR
    from sklearn.model_selection import GridSearchCV

    rng = np.random.default_rng(0)
    X = rng.standard_normal((300, 5))
    y = (X[:, 0] > 0).astype(int)
    w = rng.random(300)
    grid = {"C": [0.1, 1.0]}
    ours = DistGridSearchCV(
        SkLR(max_iter=100), grid, cv=3).fit(X, y, sample_weight=w)
    theirs = GridSearchCV(
        SkLR(max_iter=100), grid, cv=3).fit(X, y, sample_weight=w)
    np.testing.assert_allclose(
        ours.cv_results_["mean_test_score"],
        theirs.cv_results_["mean_test_score"], atol=1e-14)
    assert ours.best_index_ == theirs.best_index_


def test_fit_params_passthrough_host_estimator():
    """Arbitrary estimator-specific fit kwargs travel through the task
    fan-out to every worker fit (reference test_search.py:86-101 did
    this with xgboost eval_set/early_stopping)."""
    from sklearn.ensemble import GradientBoostingClassifier

    X = np.array([[1, 1, 1], [0, 0, 0], [-1, -1, -1]] * 60, dtype=float)
    y = np.array([0, 0, 1] * 60)
    seen = []

    def monitor(i, est, locals_):  # sklearn GBT's fit callback kwarg
        seen.append(i)
        return False

    gs = DistRandomizedSearchCV(
        GradientBoostingClassifier(n_estimators=5, random_state=0),
        {"max_depth": [2, 3]}, cv=3, n_iter=2, random_state=0,
    )
    gs.fit(X, y, monitor=monitor)
    assert np.allclose(gs.predict(X[:3]), [0, 0, 1])
    assert len(seen) > 0  # the kwarg reached the worker fits
