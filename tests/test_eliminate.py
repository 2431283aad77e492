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


def test_batched_eliminator_masked_solve():
    """Batched device path (Cluster, eager on CPU): one masked solve
    scores every (subset x fold); junk features eliminated, pickle-safe,
    agrees with the per-task generic path."""
    import pickle

    from skdist_amd import Cluster
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = 4000, 12
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = np.zeros(f)
    w[:6] = rng.standard_normal(6) * 2
    y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)
    X[:, 6:] = rng.standard_normal((n, 6))

    el = DistFeatureEliminator(
        LogisticRegression(epochs=10, random_state=0), sc=Cluster(),
        min_features_to_select=4, step=2, cv=3)
    el.fit(X, y)
    assert set(el.best_features_) == set(range(6))
    el2 = pickle.loads(pickle.dumps(el))
    np.testing.assert_array_equal(el.predict(X), el2.predict(X))
    assert (el.predict(X) == y).mean() > 0.9
    assert len(el.scores_) >= 3

    gen = DistFeatureEliminator(
        LogisticRegression(epochs=10, random_state=0), sc=None,
        min_features_to_select=4, step=2, cv=3)
    gen.fit(X, y)
    assert set(gen.best_features_) == set(el.best_features_)


def test_masked_solve_equals_column_drop():
    """Pinning features via ColumnSpec.feat_mask must reproduce the solve
    on the physically reduced matrix (eager fp32, exact)."""
    from skdist_amd.models._sgd import (
        ColumnSpec,
        DeviceDataset,
        batched_sgd_fit,
    )

    rng = np.random.default_rng(1)
    n, f = 1000, 8
    X = rng.standard_normal((n, f)).astype(np.float32)
    y = ((X[:, :4] @ rng.standard_normal(4)) > 0).astype(np.int64)
    drop = np.array([5, 6])
    keep = np.delete(np.arange(f), drop)

    ds_full = DeviceDataset(X, y, device="cpu")
    ds_full.set_cv_partition([])
    mask = np.ones((ds_full.fa, 1), dtype=np.uint8)
    mask[drop, 0] = 0
    spec = ColumnSpec(
        "cpu", col_fold=np.array([-2]), col_class=np.array([1]),
        col_lr=np.array([0.5]), col_l2=np.array([1e-4]), feat_mask=mask)
    Wm = batched_sgd_fit(ds_full, spec, "log", 5, 256, seed=0)

    ds_red = DeviceDataset(X[:, keep], y, device="cpu")
    ds_red.set_cv_partition([])
    spec_r = ColumnSpec(
        "cpu", col_fold=np.array([-2]), col_class=np.array([1]),
        col_lr=np.array([0.5]), col_l2=np.array([1e-4]))
    Wr = batched_sgd_fit(ds_red, spec_r, "log", 5, 256, seed=0)

    np.testing.assert_allclose(
        Wm[keep, 0].numpy(), Wr[: len(keep), 0].numpy(), atol=1e-6)
    np.testing.assert_allclose(
        Wm[ds_full.intercept_row, 0].item(),
        Wr[ds_red.intercept_row, 0].item(), atol=1e-6)
    assert float(Wm[drop, 0].abs().max()) == 0.0


def test_batched_eliminator_multiclass():
    """Multiclass elimination through the masked batched path: each
    subset contributes k columns per fold (internal OvR) and the refit
    model carries the kept-feature coef block per class."""
    from skdist_amd import Cluster
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = 3000, 10
    X = rng.standard_normal((n, f)).astype(np.float32)
    W = rng.standard_normal((3, 5))
    y = (X[:, :5] @ W.T).argmax(axis=1)
    X[:, 5:] = rng.standard_normal((n, 5))
    el = DistFeatureEliminator(
        LogisticRegression(epochs=10, random_state=0), sc=Cluster(),
        min_features_to_select=3, step=2, cv=3)
    el.fit(X, y)
    assert set(range(5)) <= set(el.best_features_)
    assert (el.predict(X) == y).mean() > 0.9
    assert el.best_estimator_.coef_.shape == (3, len(el.best_features_))
