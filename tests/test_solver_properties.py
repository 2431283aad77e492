"""Structural properties of the batched SGD solver (eager path = the
numerics reference the HIP kernels are asserted against).

Column independence is THE property that makes the engine's parallel
plan sound: every (candidate × fold × class) model is one weight column,
and sharding candidates across GPUs (or packing them into one solve)
must never change any column's result.
"""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from skdist_amd.models._sgd import (
    ColumnSpec,
    DeviceDataset,
    batched_sgd_fit,
)


def _solve(ds, cols, loss, epochs=3, bs=128, momentum=0.9):
    spec = ColumnSpec(
        ds.device,
        col_fold=np.asarray([c[0] for c in cols], dtype=np.int32),
        col_class=np.asarray([c[1] for c in cols], dtype=np.int32),
        col_lr=np.asarray([c[2] for c in cols], dtype=np.float32),
        col_l2=np.asarray([c[3] for c in cols], dtype=np.float32),
        col_class2=np.asarray([c[4] for c in cols], dtype=np.int32),
    )
    return batched_sgd_fit(
        ds, spec, loss, epochs, bs, seed=0, momentum=momentum
    ).cpu().numpy()


@settings(max_examples=20, deadline=None)
@given(
    loss=st.sampled_from(["log", "hinge"]),
    n_cols=st.integers(4, 8),
    seed=st.integers(0, 10_000),
)
def test_column_independence(loss, n_cols, seed):
    """W[:, S] from a joint solve == solving subset S alone, to fp32
    matmul-rounding tolerance: no column's math ever reads another
    column.  (Bitwise equality holds at FIXED batch width — see the
    order-invariance test — but BLAS blocks different widths
    differently, so cross-width comparison carries ~1e-8 rounding.)"""
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((300, 7)).astype(np.float32)
    y = rng.integers(0, 3, 300)
    ds = DeviceDataset(X, y, device="cpu")
    fold = rng.integers(0, 3, 300).astype(np.int32)
    ds.fold_id = torch.as_tensor(fold)

    cols = []
    for _ in range(n_cols):
        cols.append((
            int(rng.integers(-2, 3)),          # fold (-2 = full data)
            int(rng.integers(0, 3)),           # target class
            float(rng.uniform(0.05, 0.5)),     # lr
            float(rng.uniform(0, 1e-2)),       # l2
            int(rng.integers(-1, 3)),          # ovo partner (-1 = none)
        ))
    W_joint = _solve(ds, cols, loss)
    pick = sorted(
        rng.choice(n_cols, size=max(2, n_cols // 2), replace=False)
    )
    W_alone = _solve(ds, [cols[i] for i in pick], loss)
    np.testing.assert_allclose(
        W_joint[:, pick], W_alone, rtol=1e-4, atol=1e-6
    )
    # 1-column subset: same math through BLAS's gemv path
    W_one = _solve(ds, [cols[pick[0]]], loss)
    np.testing.assert_allclose(
        W_joint[:, pick[0]:pick[0] + 1], W_one, rtol=1e-4, atol=1e-6
    )


@settings(max_examples=10, deadline=None)
@given(seed=st.integers(0, 10_000))
def test_column_order_invariance(seed):
    """Permuting the columns permutes the solution — nothing couples
    columns through their order."""
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((250, 5)).astype(np.float32)
    y = rng.integers(0, 2, 250)
    ds = DeviceDataset(X, y, device="cpu")
    ds.set_cv_partition([])
    cols = [
        (-2, 1, 0.3, 1e-3, -1),
        (-2, 0, 0.2, 0.0, -1),
        (-2, 1, 0.4, 1e-4, -1),
        (-2, 0, 0.1, 1e-2, -1),
    ]
    perm = list(rng.permutation(4))
    W = _solve(ds, cols, "log")
    Wp = _solve(ds, [cols[i] for i in perm], "log")
    np.testing.assert_array_equal(W[:, perm], Wp)


def test_feature_mask_column_independence():
    """Masked (feature-subset) columns solve independently too — the
    eliminator's masked batch equals per-subset solves."""
    rng = np.random.default_rng(3)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.int64)
    ds = DeviceDataset(X, y, device="cpu")
    ds.set_cv_partition([])
    fa = ds.fa
    m1 = np.ones(fa, dtype=np.uint8)
    m2 = np.ones(fa, dtype=np.uint8)
    m2[[2, 4]] = 0

    def solve(masks):
        ncols = masks.shape[1]
        spec = ColumnSpec(
            ds.device,
            col_fold=np.full(ncols, -2, dtype=np.int32),
            col_class=np.ones(ncols, dtype=np.int32),
            col_lr=np.full(ncols, 0.3, dtype=np.float32),
            col_l2=np.full(ncols, 1e-3, dtype=np.float32),
            feat_mask=masks,
        )
        return batched_sgd_fit(ds, spec, "log", 3, 128, seed=0,
                               momentum=0.9).cpu().numpy()

    W_joint = solve(np.stack([m1, m2], axis=1))
    W_m2 = solve(m2[:, None])
    np.testing.assert_allclose(W_joint[:, 1:], W_m2, rtol=1e-4, atol=1e-6)
    assert (W_joint[[2, 4], 1] == 0).all()  # masked rows pinned to zero


@settings(max_examples=15, deadline=None)
@given(
    seed=st.integers(0, 10_000),
    cv=st.integers(2, 4),
    k=st.integers(2, 3),
    grid_n=st.integers(1, 4),
)
def test_generic_search_equals_sklearn_gridsearch(seed, cv, k, grid_n):
    """The generic fan-out is sklearn's GridSearchCV, exactly: same
    mean scores, same best candidate, on random tiny problems."""
    import warnings

    from sklearn.linear_model import LogisticRegression as SkLR
    from sklearn.model_selection import GridSearchCV

    from skdist_amd.distribute.search import DistGridSearchCV

    rng = np.random.default_rng(seed)
    n = int(rng.integers(60, 200))
    f = int(rng.integers(2, 6))
    X = rng.standard_normal((n, f))
    y = rng.integers(0, k, n)
    y[: k] = np.arange(k)  # every class present
    grid = {"C": sorted(rng.uniform(0.01, 10, grid_n).tolist())}
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        ours = DistGridSearchCV(
            SkLR(solver="liblinear"), grid, cv=cv
        ).fit(X, y)
        ref = GridSearchCV(SkLR(solver="liblinear"), grid, cv=cv).fit(X, y)
    np.testing.assert_allclose(
        ours.cv_results_["mean_test_score"],
        ref.cv_results_["mean_test_score"],
    )
    assert ours.best_index_ == ref.best_index_
    np.testing.assert_allclose(ours.best_score_, ref.best_score_)
