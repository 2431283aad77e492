"""GPU tests: batched masked-solve feature eliminator + OvR device
inference (reference analogs: skdist/distribute/eliminate.py,
multiclass.py:337-362)."""

import pickle

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from skdist_amd import Cluster
    from skdist_amd.distribute.eliminate import DistFeatureEliminator
    from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
    from skdist_amd.distribute.predict import DistPredictor
    from skdist_amd.models import LogisticRegression


def test_eliminator_hip_path():
    rng = np.random.default_rng(0)
    n, f = 20000, 24
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = np.zeros(f)
    w[:12] = rng.standard_normal(12) * 2
    y = ((X @ w + 0.2 * rng.standard_normal(n)) > 0).astype(np.int64)
    X[:, 12:] = rng.standard_normal((n, 12))

    el = DistFeatureEliminator(
        LogisticRegression(epochs=12, random_state=0),
        sc=Cluster(require_gpu=True),
        min_features_to_select=8, step=4, cv=3)
    el.fit(X, y)
    # all 12 signal features kept; bf16 scoring may keep one extra rung
    assert set(range(12)) <= set(el.best_features_), el.best_features_
    assert len(el.best_features_) <= 16
    el2 = pickle.loads(pickle.dumps(el))
    np.testing.assert_array_equal(el.predict(X), el2.predict(X))
    assert (el.predict(X) == y).mean() > 0.9


def test_ovr_device_inference_matches_host():
    rng = np.random.default_rng(1)
    n, f, k = 8000, 32, 6
    X = rng.standard_normal((n, f)).astype(np.float32)
    W = rng.standard_normal((k, f))
    y = (X @ W.T).argmax(axis=1)

    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=12, random_state=0), norm="l1",
        sc=Cluster(require_gpu=True))
    ovr.fit(X, y)
    # explicit host reference (fitted model itself now scores on device)
    cols = np.column_stack(
        [est.predict_proba(X)[:, -1] for est in ovr.estimators_])
    host = cols / cols.sum(axis=1, keepdims=True)

    pred = DistPredictor(ovr, sc=None, method="predict_proba")
    dev = pred(X)
    np.testing.assert_allclose(dev, host, atol=2e-5)
    predp = DistPredictor(ovr, sc=None, method="predict")
    # fp32 GEMM vs float64 host: near-boundary rows may flip
    assert (predp(X) == ovr.predict(X)).mean() > 0.999


def test_ovo_device_inference_matches_host():
    from skdist_amd.distribute.multiclass import DistOneVsOneClassifier

    rng = np.random.default_rng(4)
    n, f, k = 6000, 24, 5
    X = rng.standard_normal((n, f)).astype(np.float32)
    W = rng.standard_normal((k, f))
    y = (X @ W.T).argmax(axis=1)
    ovo = DistOneVsOneClassifier(
        LogisticRegression(epochs=12, random_state=0),
        sc=Cluster(require_gpu=True))
    ovo.fit(X, y)
    pred = DistPredictor(ovo, sc=None, method="predict")
    assert (pred(X) == ovo.predict(X)).mean() > 0.999
