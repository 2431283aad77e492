"""
DistFeatureEliminator tests (reference mirror:
skdist/distribute/tests/test_eliminate.py).
"""

import numpy as np
import pytest
from scipy.sparse import csr_matrix
from sklearn.datasets import load_iris
from sklearn.linear_model import LogisticRegression as SkLogReg

from skdist_amd import Cluster
from skdist_amd.distribute.eliminate import DistFeatureEliminator


@pytest.fixture
def iris_with_junk():
    X, y = load_iris(return_X_y=True)
    rng = np.random.RandomState(0)
    junk = rng.uniform(size=(X.shape[0], 1)) * 0.001
    return np.hstack([X, junk]), y


def test_junk_feature_eliminated(iris_with_junk):
    X, y = iris_with_junk
    fe = DistFeatureEliminator(
        SkLogReg(solver="liblinear"), min_features_to_select=3, cv=3
    )
    fe.fit(X, y)
    assert 4 not in fe.best_features_  # junk column dropped
    assert fe.best_score_ > 0.9
    assert fe.n_features_ >= 3


def test_scores_ladder(iris_with_junk):
    X, y = iris_with_junk
    fe = DistFeatureEliminator(
        SkLogReg(solver="liblinear"), min_features_to_select=2, step=1, cv=3
    )
    fe.fit(X, y)
    assert len(fe.scores_) == 4  # remove 0,1,2,3 features
    preds = fe.predict(X)
    assert preds.shape == (len(y),)
    assert fe.predict_proba(X).shape == (len(y), 3)
    assert fe.transform(X).shape[1] == fe.n_features_


def test_sparse_input(iris_with_junk):
    X, y = iris_with_junk
    fe = DistFeatureEliminator(
        SkLogReg(solver="liblinear"), min_features_to_select=3, cv=3
    )
    fe.fit(csr_matrix(X), y)
    assert fe.best_score_ > 0.9


def test_eliminator_cluster(iris_with_junk):
    X, y = iris_with_junk
    fe = DistFeatureEliminator(
        SkLogReg(solver="liblinear"), min_features_to_select=3, cv=3,
        sc=Cluster(),
    )
    fe.fit(X, y)
    assert fe.sc is None
    assert fe.best_score_ > 0.9


def test_score_method(iris_with_junk):
    X, y = iris_with_junk
    fe = DistFeatureEliminator(
        SkLogReg(solver="liblinear"), min_features_to_select=3, cv=3
    )
    fe.fit(X, y)
    assert fe.score(X, y) > 0.9
