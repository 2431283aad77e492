"""GPU tests for the boosted family's device path (promoted from
tools/gpu_validation_r2.sh after its round-2 hardware validation run:
device GBT fit quality, class_weight through the HIP row-weight plane,
sparse-densify ingestion into a device search).

Reference workload analog: the xgboost pass-through searches of
reference test_spark.py:165-187 (boosting rides the same fan-out).
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from skdist_amd import Cluster
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import (
        HistGradientBoostingClassifier,
        HistGradientBoostingRegressor,
        LogisticRegression,
    )


def test_gbt_classifier_device_quality():
    rng = np.random.default_rng(0)
    X = rng.standard_normal((60_000, 16)).astype(np.float32)
    t = np.sin(X[:, 0]) + 0.5 * X[:, 1] ** 2 + X[:, 2]
    y = (t > np.median(t)).astype(int)
    m = HistGradientBoostingClassifier(
        n_estimators=60, random_state=0
    ).fit(X[:50_000], y[:50_000])
    acc = (m.predict(X[50_000:]) == y[50_000:]).mean()
    assert acc > 0.9, acc
    p = m.predict_proba(X[:100])
    assert p.shape == (100, 2)
    np.testing.assert_allclose(p.sum(axis=1), 1.0, atol=1e-6)


def test_gbt_regressor_device_quality():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((40_000, 12)).astype(np.float32)
    t = X[:, 0] * 2 + np.abs(X[:, 1]) + 0.1 * rng.standard_normal(len(X))
    m = HistGradientBoostingRegressor(
        n_estimators=50, random_state=0
    ).fit(X[:30_000], t[:30_000])
    pred = m.predict(X[30_000:])
    ss_res = np.sum((t[30_000:] - pred) ** 2)
    ss_tot = np.sum((t[30_000:] - t[30_000:].mean()) ** 2)
    assert 1 - ss_res / ss_tot > 0.8


def test_class_weight_through_hip_weight_plane():
    from sklearn.metrics import recall_score

    rng = np.random.default_rng(0)
    n = 100_000
    X = rng.standard_normal((n, 32)).astype(np.float32)
    y = (X[:, 0] * 2 - 2.8 + 0.5 * rng.standard_normal(n) > 0).astype(int)
    m0 = LogisticRegression(epochs=10, random_state=0).fit(X, y)
    m1 = LogisticRegression(
        epochs=10, class_weight="balanced", random_state=0
    ).fit(X, y)
    r0 = recall_score(y, m0.predict(X))
    r1 = recall_score(y, m1.predict(X))
    assert r1 > r0 + 0.05, (r0, r1)


def test_sparse_ingestion_device_search():
    import scipy.sparse as sp

    rng = np.random.default_rng(0)
    Xd = rng.standard_normal((50_000, 64)).astype(np.float32)
    Xd[Xd < 0.8] = 0
    X = sp.csr_matrix(Xd)
    y = (Xd[:, 0] + Xd[:, 1] > 0.5).astype(np.int64)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3, sc=Cluster(require_gpu=True),
    )
    gs.fit(X, y)
    assert gs.best_score_ > 0.8
