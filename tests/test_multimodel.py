"""DistMultiModelSearch tests (reference search.py:717-908)."""

import numpy as np
from scipy.stats import uniform
from sklearn.datasets import load_breast_cancer
from sklearn.ensemble import RandomForestClassifier
from sklearn.linear_model import LogisticRegression as SkLogReg

from skdist_amd import Cluster
from skdist_amd.distribute.search import DistMultiModelSearch


def _models():
    return [
        ("logreg", SkLogReg(solver="liblinear"), {"C": [0.1, 1.0, 10.0]}),
        ("rf", RandomForestClassifier(n_estimators=10, random_state=0),
         {"max_depth": [2, 4, 8]}, 2),
    ]


def test_multimodel_local():
    X, y = load_breast_cancer(return_X_y=True)
    mm = DistMultiModelSearch(
        _models(), n=3, cv=3, random_state=0, scoring="roc_auc"
    )
    mm.fit(X, y)
    assert mm.best_score_ > 0.95
    assert mm.best_model_name_ in ("logreg", "rf")
    assert mm.worst_score_ <= mm.best_score_
    r = mm.cv_results_
    assert len(r["params"]) == len(r["mean_test_score"])
    assert set(r["model_name"]) == {"logreg", "rf"}
    # rf capped at n=2 samples, logreg n=3
    assert r["model_name"].count("rf") == 2
    preds = mm.predict(X[:5])
    assert preds.shape == (5,)


def test_multimodel_cluster():
    X, y = load_breast_cancer(return_X_y=True)
    mm = DistMultiModelSearch(
        _models(), n=2, cv=3, random_state=0, sc=Cluster()
    )
    mm.fit(X, y)
    assert mm.sc is None
    assert mm.best_score_ > 0.9


def test_multimodel_continuous_dists():
    X, y = load_breast_cancer(return_X_y=True)
    mm = DistMultiModelSearch(
        [("lr", SkLogReg(solver="liblinear"), {"C": uniform(0.1, 5)})],
        n=3, cv=3, random_state=7,
    )
    mm.fit(X, y)
    assert len(mm.cv_results_["params"]) == 3


def test_multimodel_batched_path_matches_generic():
    """With a Cluster, our linear families solve all their sampled
    param sets as one batched solve per model; results must agree with
    the generic per-task path."""
    from skdist_amd import Cluster
    from skdist_amd.models import LinearSVC, LogisticRegression

    rng = np.random.default_rng(0)
    X = rng.standard_normal((1200, 10)).astype(np.float32)
    y = ((X[:, :4] @ rng.standard_normal(4)) > 0).astype(np.int64)
    models = [
        ("lr", LogisticRegression(epochs=10, random_state=0),
         {"C": [0.1, 1.0, 10.0]}),
        ("svc", LinearSVC(epochs=10, random_state=0),
         {"C": [0.1, 1.0]}),
    ]
    mm_b = DistMultiModelSearch(models, n=2, cv=3, sc=Cluster(),
                                random_state=0)
    mm_b.fit(X, y)
    mm_g = DistMultiModelSearch(models, n=2, cv=3, sc=None,
                                random_state=0)
    mm_g.fit(X, y)
    assert mm_b.best_model_name_ in ("lr", "svc")
    assert len(mm_b.cv_results_["params"]) == len(
        mm_g.cv_results_["params"])
    # same sampled candidates, closely matching scores
    assert mm_b.cv_results_["params"] == mm_g.cv_results_["params"]
    np.testing.assert_allclose(
        mm_b.cv_results_["mean_test_score"],
        mm_g.cv_results_["mean_test_score"], atol=0.05)
    assert mm_b.best_score_ > 0.8


def test_multimodel_with_boosted_family():
    """Heterogeneous pool mixing the native linear and boosted families
    (reference's pool mixed sklearn + xgboost, search.py:717-908)."""
    from skdist_amd.models import (
        HistGradientBoostingClassifier,
        LogisticRegression,
    )

    rng = np.random.default_rng(0)
    X = rng.standard_normal((600, 6)).astype(np.float32)
    y = (np.sin(X[:, 0]) + X[:, 1] > 0).astype(int)
    mm = DistMultiModelSearch(
        [
            ("lr", LogisticRegression(epochs=8, random_state=0),
             {"C": [0.1, 1.0]}),
            ("gbt", HistGradientBoostingClassifier(
                n_estimators=20, random_state=0),
             {"max_depth": [2, 3]}),
        ],
        n=2, cv=3, random_state=0,
    )
    mm.fit(X, y)
    assert mm.best_score_ > 0.85
    assert mm.best_model_name_ in ("lr", "gbt")
    assert mm.predict(X[:4]).shape == (4,)
