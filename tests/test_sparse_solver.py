"""CPU tests for the sparse-native batched linear solver
(skdist_amd/models/_sparse_sgd.py) — the text-scale path.

Equivalence target: the dense batched solver with standardize=False and
momentum=0 runs the mathematically identical update (the sparse path's
lazy L2 scale is exactly the dense per-step decay; see the kernel
header), so coefficients must agree to bf16-gradient tolerance.

Reference workloads: sk-dist's HashingVectorizer -> LR/SVC pipelines
(reference _defaults.py:91-198, examples/postprocessing/simple_voter.py).
"""

import os

import numpy as np
import pytest
import scipy.sparse as sp

FORCE = {"SKDIST_AMD_FORCE_SPARSE": "1"}


@pytest.fixture()
def force_sparse(monkeypatch):
    monkeypatch.setenv("SKDIST_AMD_FORCE_SPARSE", "1")


def _data(n=3000, f=200, seed=0, sparsity=1.0):
    rng = np.random.default_rng(seed)
    Xd = rng.standard_normal((n, f)).astype(np.float32)
    Xd[np.abs(Xd) < sparsity] = 0
    w = rng.standard_normal(f) * (rng.random(f) < 0.2)
    y = ((Xd @ w + 0.3 * rng.standard_normal(n)) > 0).astype(int)
    return Xd, sp.csr_matrix(Xd), y, w


def test_sparse_matches_dense_solver(force_sparse):
    from skdist_amd.models import LogisticRegression

    Xd, X, y, _ = _data()
    kw = dict(epochs=20, momentum=0.0, random_state=0, adaptive=False)
    m_sp = LogisticRegression(**kw).fit(X, y)
    os.environ.pop("SKDIST_AMD_FORCE_SPARSE")
    m_de = LogisticRegression(standardize=False, **kw).fit(Xd, y)
    # same update math; sparse rounds the per-batch gradients to bf16
    corr = np.corrcoef(m_sp.coef_[0], m_de.coef_[0])[0, 1]
    assert corr > 0.999, corr
    np.testing.assert_allclose(
        m_sp.intercept_, m_de.intercept_, atol=5e-3)
    agree = (m_sp.predict(Xd) == m_de.predict(Xd)).mean()
    assert agree > 0.995, agree


def test_sparse_deterministic(force_sparse):
    from skdist_amd.models import LinearSVC

    _, X, y, _ = _data(seed=3)
    m1 = LinearSVC(epochs=10, momentum=0.0, random_state=0).fit(X, y)
    m2 = LinearSVC(epochs=10, momentum=0.0, random_state=0).fit(X, y)
    np.testing.assert_array_equal(m1.coef_, m2.coef_)
    np.testing.assert_array_equal(m1.intercept_, m2.intercept_)


def test_sparse_search_scores_match_dense(force_sparse):
    from sklearn.model_selection import KFold

    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    Xd, X, y, _ = _data(n=2400, f=120, seed=1)
    cv = KFold(3)
    grid = {"C": [0.1, 1.0, 10.0]}
    kw = dict(epochs=12, momentum=0.0, random_state=0, adaptive=False)
    g_sp = DistGridSearchCV(
        LogisticRegression(**kw), grid, cv=cv).fit(X, y)
    os.environ.pop("SKDIST_AMD_FORCE_SPARSE")
    g_de = DistGridSearchCV(
        LogisticRegression(standardize=False, **kw), grid, cv=cv
    ).fit(Xd, y)
    np.testing.assert_allclose(
        g_sp.cv_results_["mean_test_score"],
        g_de.cv_results_["mean_test_score"], atol=6e-3)
    assert g_sp.best_score_ > 0.85


@pytest.mark.parametrize("metric", ["roc_auc", "f1", "neg_log_loss"])
def test_sparse_device_metrics(force_sparse, metric):
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    _, X, y, _ = _data(n=1800, f=80, seed=2)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, momentum=0.0, random_state=0),
        {"C": [0.5, 2.0]}, cv=3, scoring=metric).fit(X, y)
    assert np.isfinite(gs.best_score_)
    if metric == "roc_auc":
        assert gs.best_score_ > 0.9


def test_sparse_multiclass_ovr_ovo(force_sparse):
    from skdist_amd.distribute.multiclass import (
        DistOneVsOneClassifier,
        DistOneVsRestClassifier,
    )
    from skdist_amd.models import LinearSVC, LogisticRegression

    rng = np.random.default_rng(4)
    n, f, k = 2400, 150, 3
    Xd = rng.standard_normal((n, f)).astype(np.float32)
    Xd[np.abs(Xd) < 1.0] = 0
    X = sp.csr_matrix(Xd)
    y = (Xd @ rng.standard_normal((f, k))).argmax(axis=1)
    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=12, momentum=0.0, random_state=0)
    ).fit(X, y)
    assert (ovr.predict(X) == y).mean() > 0.85
    ovo = DistOneVsOneClassifier(
        LinearSVC(epochs=12, momentum=0.0, random_state=0)
    ).fit(X, y)
    assert (ovo.predict(X) == y).mean() > 0.85


def test_sparse_ridge_and_sample_weight(force_sparse):
    from skdist_amd.models import Ridge

    rng = np.random.default_rng(5)
    Xd, X, y, w = _data(n=2000, f=100, seed=5)
    t = (Xd @ w).astype(np.float64)
    r = Ridge(epochs=20, random_state=0).fit(X, t)
    pred = r.predict(Xd)
    assert 1 - np.sum((t - pred) ** 2) / np.sum(
        (t - t.mean()) ** 2) > 0.99
    # sample_weight flows through the fused row-weight plane
    from skdist_amd.models import LogisticRegression

    sw = rng.random(len(y)).astype(np.float32) + 0.5
    m_sp = LogisticRegression(
        epochs=15, momentum=0.0, random_state=0, adaptive=False
    ).fit(X, y, sample_weight=sw)
    os.environ.pop("SKDIST_AMD_FORCE_SPARSE")
    m_de = LogisticRegression(
        epochs=15, momentum=0.0, standardize=False, random_state=0
    ).fit(Xd, y, sample_weight=sw)
    assert np.corrcoef(m_sp.coef_[0], m_de.coef_[0])[0, 1] > 0.999


def test_sparse_lazy_l2_matches_direct_decay(force_sparse):
    """Heavy regularization (small C): the lazy scale must equal the
    dense path's direct per-step decay, including through a renorm."""
    from skdist_amd.models import LogisticRegression

    Xd, X, y, _ = _data(n=1500, f=60, seed=6)
    kw = dict(C=1e-3, epochs=20, lr=1.0, momentum=0.0, random_state=0,
              adaptive=False)
    m_sp = LogisticRegression(**kw).fit(X, y)
    os.environ.pop("SKDIST_AMD_FORCE_SPARSE")
    m_de = LogisticRegression(standardize=False, **kw).fit(Xd, y)
    np.testing.assert_allclose(
        m_sp.coef_, m_de.coef_, rtol=0.05, atol=1e-4)
    np.testing.assert_allclose(
        m_sp.intercept_, m_de.intercept_, atol=5e-3)


def test_widefeature_text_flow_auto_sparse():
    """f = 2^20 routes to the sparse path without any env override and
    completes an end-to-end search (the reference's hashed-text shape)."""
    from sklearn.feature_extraction.text import HashingVectorizer

    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    v0 = [f"tok{i}" for i in range(300)]
    v1 = [f"tok{i}" for i in range(200, 500)]
    docs = []
    y = np.arange(800) % 2
    for c in y:
        docs.append(" ".join(rng.choice(v1 if c else v0, size=20)))
    X = HashingVectorizer(n_features=2 ** 20).transform(docs)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=8, momentum=0.0, random_state=0),
        {"C": [1.0]}, cv=3).fit(X, y)
    assert gs.best_score_ > 0.9
    assert gs.best_estimator_.coef_.shape == (1, 2 ** 20)


def test_sparse_momentum_warns_once(force_sparse):
    import warnings

    from skdist_amd.models import LogisticRegression

    _, X, y, _ = _data(n=600, f=40, seed=7)
    with warnings.catch_warnings(record=True) as rec:
        warnings.simplefilter("always")
        LogisticRegression(epochs=3, random_state=0).fit(X, y)  # momentum=.9
    assert any("momentum-free" in str(r.message) for r in rec)
