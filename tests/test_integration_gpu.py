"""GPU integration tests for composed flows that previously only ran
on CPU: Encoderizer text tiers feeding a device search, and
DistMultiModelSearch sharding native models on the GPU cluster."""

import numpy as np
import pandas as pd
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from skdist_amd import Cluster
    from skdist_amd.distribute.encoder import Encoderizer
    from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
    from skdist_amd.distribute.search import DistMultiModelSearch
    from skdist_amd.models import (
        HistGradientBoostingClassifier,
        LinearSVC,
        LogisticRegression,
    )


def _mixed_text_frame(n=4000, seed=0):
    rng = np.random.default_rng(seed)
    v0 = [f"alpha{i}" for i in range(300)]
    v1 = [f"omega{i}" for i in range(300)]
    y = (rng.random(n) < 0.5).astype(np.int64)
    docs = [
        " ".join(rng.choice(v1 if c else v0, size=15)) for c in y
    ]
    df = pd.DataFrame({
        "text": docs,
        "num": rng.standard_normal(n),
        "color": [["red", "blue", "lime"][i]
                  for i in rng.integers(0, 3, size=n)],
    })
    return df, y


def test_encoderizer_text_flow_to_device_ovr():
    """Encoderizer('small') mixed frame (device hashing vectorizer for
    the text column) -> native OvR on the GPU cluster, end to end."""
    df, y = _mixed_text_frame()
    enc = Encoderizer(size="small", sc=Cluster(require_gpu=True))
    Xt = enc.fit_transform(df)
    assert Xt.shape[0] == len(df)
    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=10, momentum=0.0, random_state=0),
        sc=Cluster(require_gpu=True),
    ).fit(Xt, y)
    acc = (ovr.predict(enc.transform(df)) == y).mean()
    assert acc > 0.95, acc


def test_multimodel_search_native_models_on_device():
    """Heterogeneous model pool (native LR / SVC / hist-GBT) sharded on
    the GPU cluster (reference DistMultiModelSearch analog)."""
    rng = np.random.default_rng(1)
    n, f = 20_000, 24
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f)
    y = ((X @ w + 0.4 * rng.standard_normal(n)) > 0).astype(np.int64)
    models = [
        ("lr", LogisticRegression(epochs=8, random_state=0),
         {"C": [0.1, 1.0]}),
        ("svc", LinearSVC(epochs=8, random_state=0), {"C": [1.0]}),
        ("gbt", HistGradientBoostingClassifier(
            n_estimators=20, random_state=0),
         {"learning_rate": [0.1, 0.2]}),
    ]
    ms = DistMultiModelSearch(
        models, cv=3, n=2, random_state=0,
        sc=Cluster(require_gpu=True))
    ms.fit(X, y)
    assert ms.best_estimator_ is not None
    assert (ms.predict(X[:200]) == y[:200]).mean() > 0.7


def test_ovr_text_serving_sparse_device():
    """Fitted OvR over hashed text served through DistPredictor with
    CSR input: the device hook's sparse branch must match host scoring."""
    import scipy.sparse as sp

    from skdist_amd.distribute.predict import DistPredictor

    rng = np.random.default_rng(3)
    n, f, k = 30_000, 1 << 17, 4
    rows = np.repeat(np.arange(n), 15)
    cols = np.sort(rng.integers(0, f, size=(n, 15)), axis=1).ravel()
    vals = np.full(n * 15, 0.26, dtype=np.float32)
    X = sp.csr_matrix((vals, (rows, cols.astype(np.int64))),
                      shape=(n, f))
    W = np.zeros((f, k), dtype=np.float32)
    W[: 4096] = rng.standard_normal((4096, k))
    y = np.asarray(X @ W).argmax(axis=1)
    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=8, momentum=0.0, random_state=0),
        sc=Cluster(require_gpu=True)).fit(X, y)
    pred = DistPredictor(ovr, sc=Cluster(require_gpu=True),
                         method="predict")
    p_dev = pred(X[:4000])
    p_host_fn = getattr(ovr, "predict")
    agree = (p_dev == p_host_fn(X[:4000])).mean()
    assert agree > 0.999, agree
