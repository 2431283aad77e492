"""Native hist-GBT family (models/boosting.py): quality vs sklearn's
GradientBoosting, sklearn-API behavior, and riding the meta-estimators
(the reference's boosted workloads were an xgboost pass-through,
reference README.rst:158-166 / test_spark.py:165-187 — this engine ships
its own boosted family on the binned tree builder instead)."""

import pickle

import numpy as np
import pytest
from sklearn.ensemble import (
    GradientBoostingClassifier,
    GradientBoostingRegressor,
)
from sklearn.model_selection import train_test_split

from skdist_amd.models import (
    HistGradientBoostingClassifier,
    HistGradientBoostingRegressor,
)


@pytest.fixture(scope="module")
def data():
    rng = np.random.default_rng(0)
    X = rng.standard_normal((3000, 10)).astype(np.float32)
    t = (np.sin(X[:, 0]) + 0.5 * X[:, 1] ** 2 + X[:, 2]).astype(np.float64)
    return X, t


def _r2(m, X, y):
    p = m.predict(X)
    return 1 - ((y - p) ** 2).sum() / ((y - y.mean()) ** 2).sum()


def test_regressor_quality_vs_sklearn(data):
    X, t = data
    Xtr, Xte, ttr, tte = train_test_split(X, t, random_state=0)
    ours = HistGradientBoostingRegressor(
        n_estimators=60, random_state=0
    ).fit(Xtr, ttr)
    ref = GradientBoostingRegressor(n_estimators=60, random_state=0).fit(
        Xtr, ttr
    )
    assert _r2(ours, Xte, tte) > _r2(ref, Xte, tte) - 0.03
    assert _r2(ours, Xte, tte) > 0.9


def test_binary_classifier_quality(data):
    X, t = data
    y = (t > np.median(t)).astype(int)
    Xtr, Xte, ytr, yte = train_test_split(X, y, random_state=0)
    ours = HistGradientBoostingClassifier(
        n_estimators=60, random_state=0
    ).fit(Xtr, ytr)
    ref = GradientBoostingClassifier(n_estimators=60, random_state=0).fit(
        Xtr, ytr
    )
    acc_o = (ours.predict(Xte) == yte).mean()
    assert acc_o > (ref.predict(Xte) == yte).mean() - 0.03
    p = ours.predict_proba(Xte)
    assert p.shape == (len(yte), 2)
    assert np.allclose(p.sum(axis=1), 1.0)
    # probabilities are informative, not just 0/1
    assert 0.05 < p[:, 1].std() < 0.5
    assert ours.decision_function(Xte).shape == (len(yte),)


def test_multiclass_classifier_quality(data):
    X, t = data
    y = np.digitize(t, np.quantile(t, [0.33, 0.66]))
    Xtr, Xte, ytr, yte = train_test_split(X, y, random_state=0)
    ours = HistGradientBoostingClassifier(
        n_estimators=40, random_state=0
    ).fit(Xtr, ytr)
    ref = GradientBoostingClassifier(n_estimators=40, random_state=0).fit(
        Xtr, ytr
    )
    assert (ours.predict(Xte) == yte).mean() > (
        (ref.predict(Xte) == yte).mean() - 0.03
    )
    p = ours.predict_proba(Xte)
    assert p.shape == (len(yte), 3)
    assert np.allclose(p.sum(axis=1), 1.0)


def test_subsample_string_labels_pickle(data):
    X, t = data
    y = np.array(["lo", "mid", "hi"])[
        np.digitize(t, np.quantile(t, [0.33, 0.66]))
    ]
    m = HistGradientBoostingClassifier(
        n_estimators=30, subsample=0.6, random_state=0
    ).fit(X, y)
    assert list(m.classes_) == ["hi", "lo", "mid"]
    back = pickle.loads(pickle.dumps(m))
    assert (back.predict(X) == m.predict(X)).all()
    assert (m.predict(X) == y).mean() > 0.85
    # deterministic under a fixed seed
    m2 = HistGradientBoostingClassifier(
        n_estimators=30, subsample=0.6, random_state=0
    ).fit(X, y)
    assert (m2.predict(X) == m.predict(X)).all()


def test_estimators_shape_and_importances(data):
    X, t = data
    m = HistGradientBoostingRegressor(
        n_estimators=10, random_state=0
    ).fit(X, t)
    assert m.estimators_.shape == (10, 1)
    imp = m.feature_importances_
    assert imp.shape == (10,)
    assert np.isclose(imp.sum(), 1.0)
    # features 0..2 carry the signal
    assert imp[:3].sum() > 0.8


def test_rides_search_and_predictor(data):
    X, t = data
    y = (t > np.median(t)).astype(int)
    from skdist_amd.distribute.predict import DistPredictor
    from skdist_amd.distribute.search import DistGridSearchCV

    gs = DistGridSearchCV(
        HistGradientBoostingClassifier(n_estimators=15, random_state=0),
        {"max_depth": [2, 3], "learning_rate": [0.1, 0.3]}, cv=3,
    )
    gs.fit(X, y)
    assert gs.best_score_ > 0.85
    assert set(gs.best_params_) == {"max_depth", "learning_rate"}
    blob = pickle.loads(pickle.dumps(gs))
    pred = DistPredictor(blob.best_estimator_, method="predict_proba")
    out = pred(X)
    assert np.allclose(out, blob.best_estimator_.predict_proba(X))


def test_early_stopping(data):
    X, t = data
    y = (t > np.median(t)).astype(int)
    m = HistGradientBoostingClassifier(
        n_estimators=200, n_iter_no_change=5, random_state=0
    ).fit(X, y)
    assert m.n_estimators_ < 200
    assert (m.predict(X) == y).mean() > 0.9
    r = HistGradientBoostingRegressor(
        n_estimators=300, n_iter_no_change=5, random_state=0
    ).fit(X, X[:, 0] * 2.0)
    assert r.n_estimators_ < 300


def test_staged_predictions(data):
    """staged_* iterators (sklearn protocol): stage i equals a model
    trained with i+1 rounds; the last stage equals predict."""
    X, t = data
    y = (t > np.median(t)).astype(int)
    m = HistGradientBoostingClassifier(
        n_estimators=8, random_state=0
    ).fit(X, y)
    stages = list(m.staged_predict_proba(X))
    assert len(stages) == 8
    np.testing.assert_allclose(stages[-1], m.predict_proba(X))
    m3 = HistGradientBoostingClassifier(
        n_estimators=3, random_state=0
    ).fit(X, y)
    np.testing.assert_allclose(stages[2], m3.predict_proba(X), atol=1e-6)
    preds = list(m.staged_predict(X))
    assert (preds[-1] == m.predict(X)).all()
    r = HistGradientBoostingRegressor(n_estimators=5, random_state=0).fit(
        X, t
    )
    rs = list(r.staged_predict(X))
    assert len(rs) == 5
    np.testing.assert_allclose(rs[-1], r.predict(X))
