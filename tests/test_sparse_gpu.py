"""GPU tests for the sparse text-scale solver kernels
(ops/csrc/sparse_sgd_kernels.hip) against the eager torch mirror
(_sparse_sgd._sparse_sgd_eager — same number flow: fp32 accumulate,
bf16 G, lazy L2 scale)."""

import numpy as np
import pytest
import scipy.sparse as sp
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from skdist_amd.models._sgd import ColumnSpec
    from skdist_amd.models._sparse_sgd import (
        SparseDeviceDataset,
        _sp_forward_eager,
        sparse_sgd_fit,
    )


def _ds_and_spec(n=5000, f=400, ncols=24, seed=0, with_rw=False,
                 folds=3):
    rng = np.random.default_rng(seed)
    Xd = rng.standard_normal((n, f)).astype(np.float32)
    Xd[np.abs(Xd) < 1.1] = 0
    X = sp.csr_matrix(Xd)
    w = rng.standard_normal(f) * (rng.random(f) < 0.2)
    y = ((Xd @ w) > 0).astype(np.int64)
    rw = (rng.random(n).astype(np.float32) + 0.5) if with_rw else None
    ds = SparseDeviceDataset(X, y, sample_weight=rw)
    fold = np.arange(n) % folds
    splits = [
        (np.flatnonzero(fold != k), np.flatnonzero(fold == k))
        for k in range(folds)
    ]
    assert ds.set_cv_partition(splits)
    cf = np.tile(np.arange(folds, dtype=np.int32), ncols // folds)
    spec = ColumnSpec(
        ds.device,
        col_fold=cf,
        col_class=np.ones(ncols, dtype=np.int32),
        col_lr=np.full(ncols, 0.3, dtype=np.float32),
        col_l2=np.full(ncols, 1e-4, dtype=np.float32),
    )
    return ds, spec, X, y


def test_hip_solve_matches_eager_mirror():
    ds, spec, _, _ = _ds_and_spec()
    W_hip = sparse_sgd_fit(ds, spec, "log", epochs=5, batch_size=1024,
                           seed=0).cpu().numpy()
    W_eag = sparse_sgd_fit(ds, spec, "log", epochs=5, batch_size=1024,
                           seed=0, force_eager=True).cpu().numpy()
    # same math, different fp32 summation order
    corr = np.corrcoef(W_hip.ravel(), W_eag.ravel())[0, 1]
    assert corr > 0.9999, corr
    np.testing.assert_allclose(W_hip, W_eag, atol=2e-2, rtol=0.05)


def test_hip_solve_with_row_weights_matches_eager():
    ds, spec, _, _ = _ds_and_spec(with_rw=True, seed=1)
    W_hip = sparse_sgd_fit(ds, spec, "hinge", epochs=4, batch_size=1024,
                           seed=0).cpu().numpy()
    W_eag = sparse_sgd_fit(ds, spec, "hinge", epochs=4, batch_size=1024,
                           seed=0, force_eager=True).cpu().numpy()
    assert np.corrcoef(W_hip.ravel(), W_eag.ravel())[0, 1] > 0.999


def test_hip_solve_deterministic():
    ds, spec, _, _ = _ds_and_spec(seed=2)
    W1 = sparse_sgd_fit(ds, spec, "log", epochs=3, batch_size=1024,
                        seed=0).cpu().numpy()
    W2 = sparse_sgd_fit(ds, spec, "log", epochs=3, batch_size=1024,
                        seed=0).cpu().numpy()
    np.testing.assert_array_equal(W1, W2)


def test_sp_forward_matches_eager():
    from skdist_amd.ops import require_hip

    ext = require_hip()
    ds, spec, _, _ = _ds_and_spec(seed=3)
    dev = ds.device
    cp = 64
    rng = np.random.default_rng(0)
    Wt = torch.as_tensor(
        rng.standard_normal((ds.f, cp)).astype(np.float32), device=dev
    ).contiguous()
    Wb = torch.as_tensor(
        rng.standard_normal(cp).astype(np.float32), device=dev)
    s = torch.ones(cp, dtype=torch.float32, device=dev)
    rows = torch.arange(0, ds.n, 7, dtype=torch.int64, device=dev)
    Z_hip = torch.empty(len(rows), cp, dtype=torch.float32, device=dev)
    ext.sp_forward(ds.crow, ds.cidx, ds.cval, Wt, Wb, s, rows, Z_hip)
    Z_eag = torch.empty_like(Z_hip)
    _sp_forward_eager(ds, Wt, Wb, rows, Z_eag)
    np.testing.assert_allclose(
        Z_hip.cpu().numpy(), Z_eag.cpu().numpy(), atol=1e-3, rtol=1e-4)


def test_text_scale_search_on_device():
    """Hashed text at 2^20 features through DistGridSearchCV on the HIP
    sparse path (the VERDICT round-2 target workload, scaled to test
    size)."""
    from sklearn.feature_extraction.text import HashingVectorizer

    from skdist_amd import Cluster
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    v0 = [f"tok{i}" for i in range(2000)]
    v1 = [f"tok{i}" for i in range(1500, 3500)]
    n = 20000
    y = np.arange(n) % 2
    docs = [
        " ".join(rng.choice(v1 if c else v0, size=30)) for c in y
    ]
    X = HashingVectorizer(n_features=2 ** 20).transform(docs)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=8, momentum=0.0, random_state=0),
        {"C": [0.5, 2.0]}, cv=3, scoring="accuracy",
        sc=Cluster(require_gpu=True))
    gs.fit(X, y)
    assert gs.best_score_ > 0.9, gs.best_score_
    assert gs.best_estimator_.coef_.shape == (1, 2 ** 20)


def test_hip_heavy_l2_renorm_matches_eager():
    """Heavy regularization drives the lazy scale through the renorm
    path; HIP (host-mirrored scale, no syncs) must track the eager
    mirror."""
    ds, spec0, _, _ = _ds_and_spec(seed=4)
    # decay 0.92/step x 100 steps -> s ~ 2e-4 < 1e-3: renorm fires
    spec = ColumnSpec(
        ds.device,
        col_fold=spec0.col_fold.cpu().numpy(),
        col_class=spec0.col_class.cpu().numpy(),
        col_lr=np.full(spec0.ncols, 1.0, dtype=np.float32),
        col_l2=np.full(spec0.ncols, 0.08, dtype=np.float32),
    )
    W_hip = sparse_sgd_fit(ds, spec, "log", epochs=10, batch_size=512,
                           seed=0).cpu().numpy()
    W_eag = sparse_sgd_fit(ds, spec, "log", epochs=10, batch_size=512,
                           seed=0, force_eager=True).cpu().numpy()
    assert np.isfinite(W_hip).all()
    assert np.corrcoef(W_hip.ravel(), W_eag.ravel())[0, 1] > 0.99


def test_linear_device_serving_sparse_and_dense():
    """DistPredictor serves a fitted wide-coef LR through the device
    GEMM / sparse forward kernel; predictions match the host path."""
    from skdist_amd import Cluster
    from skdist_amd.distribute.predict import DistPredictor
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(0)
    n, f = 30_000, 1 << 17
    rows = np.repeat(np.arange(n), 20)
    cols = np.sort(rng.integers(0, f, size=len(rows)).reshape(n, 20),
                   axis=1).ravel()
    vals = np.full(len(rows), 0.22, dtype=np.float32)
    X = sp.csr_matrix((vals, (rows, cols.astype(np.int64))),
                      shape=(n, f))
    w = np.zeros(f, dtype=np.float32)
    w[: 4096] = rng.standard_normal(4096)
    y = (np.asarray(X @ w).ravel() > 0).astype(np.int64)
    m = LogisticRegression(epochs=6, momentum=0.0, random_state=0).fit(
        X, y)
    pred = DistPredictor(m, sc=Cluster(require_gpu=True),
                         method="predict_proba")
    p_dev = pred(X[:5000])
    p_host = m.predict_proba(X[:5000])
    np.testing.assert_allclose(p_dev, p_host, atol=2e-3)
    # dense chunk too
    pred2 = DistPredictor(m, sc=Cluster(require_gpu=True),
                          method="predict")
    Xd = np.asarray(X[:2000].todense(), dtype=np.float32)
    assert (pred2(Xd) == m.predict(Xd)).mean() > 0.999
