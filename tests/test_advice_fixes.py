"""Regression tests for the round-1 advisor findings:

1. boosting random_state=None must draw fresh entropy per fit;
2. HGB classes accept ``max_iter`` as an ``n_estimators`` alias and are
   exported under GradientBoosting* names too;
3. class_weight='balanced' on the batched linear path falls back to the
   generic (exact-sklearn) path when folds are not stratified;
4. get_oof aligns fold probabilities to the global class set even when a
   training fold misses a class;
5. Cluster.run_tasks wraps task failures in TaskFailedError identically
   in world-1 and distributed mode.
"""

import numpy as np
import pytest


def _xor_data(n=400, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, 6)).astype(np.float32)
    y = ((X[:, 0] > 0) ^ (X[:, 1] > 0)).astype(int)
    return X, y


def test_boosting_random_state_none_is_fresh_entropy():
    from skdist_amd.models import HistGradientBoostingClassifier

    X, y = _xor_data()
    m1 = HistGradientBoostingClassifier(
        n_estimators=5, subsample=0.5, random_state=None
    ).fit(X, y)
    m2 = HistGradientBoostingClassifier(
        n_estimators=5, subsample=0.5, random_state=None
    ).fit(X, y)
    d1 = m1.decision_function(X)
    d2 = m2.decision_function(X)
    assert not np.array_equal(d1, d2), "None seed must not be deterministic"
    # int seed stays deterministic
    m3 = HistGradientBoostingClassifier(
        n_estimators=5, subsample=0.5, random_state=7
    ).fit(X, y)
    m4 = HistGradientBoostingClassifier(
        n_estimators=5, subsample=0.5, random_state=7
    ).fit(X, y)
    np.testing.assert_array_equal(
        m3.decision_function(X), m4.decision_function(X)
    )


def test_boosting_max_iter_alias_and_gb_names():
    from skdist_amd.models import (
        GradientBoostingClassifier,
        GradientBoostingRegressor,
        HistGradientBoostingClassifier,
        HistGradientBoostingRegressor,
    )

    assert GradientBoostingClassifier is HistGradientBoostingClassifier
    assert GradientBoostingRegressor is HistGradientBoostingRegressor
    X, y = _xor_data()
    m = HistGradientBoostingClassifier(max_iter=4, random_state=0).fit(X, y)
    assert m.n_estimators_ == 4
    # max_iter participates in get_params/clone (grid-searchable)
    assert "max_iter" in m.get_params()


def test_balanced_class_weight_non_stratified_falls_back():
    from skdist_amd.models import LogisticRegression
    from skdist_amd.models.linear import FallbackToGeneric

    rng = np.random.default_rng(0)
    n = 300
    y = (rng.random(n) < 0.2).astype(int)
    order = np.argsort(y)  # label-sorted → wildly non-stratified folds
    folds = [
        (order[100:], order[:100]),
        (np.concatenate([order[:100], order[200:]]), order[100:200]),
        (order[:200], order[200:]),
    ]
    est = LogisticRegression(class_weight="balanced")
    with pytest.raises(FallbackToGeneric):
        est._check_balanced_foldable(y, folds)
    # stratified folds pass the guard
    from sklearn.model_selection import StratifiedKFold

    skf = list(StratifiedKFold(3, shuffle=True, random_state=0).split(
        np.zeros((n, 1)), y))
    est._check_balanced_foldable(y, skf)


def test_balanced_search_per_fold_weights_on_sorted_labels():
    """End-to-end: non-stratified KFold on sorted labels + 'balanced'
    takes the generic path, whose per-fold fits recompute 'balanced'
    from each training fold (sklearn semantics) — asserted against a
    hand-rolled per-fold loop with the same estimator."""
    from sklearn.model_selection import KFold

    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    rng = np.random.default_rng(1)
    n = 240
    X = rng.standard_normal((n, 5)).astype(np.float32)
    y = np.concatenate([np.zeros(140, int), np.ones(100, int)])
    X[y == 1, 0] += 2.0
    cv = list(KFold(3).split(X))  # sorted labels → wildly non-stratified
    grid = {"C": [0.1, 1.0]}
    base = LogisticRegression(
        class_weight="balanced", epochs=10, random_state=0
    )
    ours = DistGridSearchCV(base, grid, cv=cv).fit(X, y)

    expect = []
    for C in grid["C"]:
        scores = []
        for tr, te in cv:
            est = LogisticRegression(
                C=C, class_weight="balanced", epochs=10, random_state=0
            ).fit(X[tr], y[tr])
            scores.append((est.predict(X[te]) == y[te]).mean())
        expect.append(np.mean(scores))
    np.testing.assert_allclose(
        ours.cv_results_["mean_test_score"], expect, rtol=1e-6
    )


def test_get_oof_with_fold_missing_a_class():
    from sklearn.linear_model import LogisticRegression as SkLR

    from skdist_amd.distribute.ensemble import get_oof

    rng = np.random.default_rng(0)
    n = 150
    X = rng.standard_normal((n, 4))
    y = np.repeat([0, 1, 2], n // 3)  # label-sorted: unshuffled KFold
    # folds will miss classes in training splits
    fitted, oof = get_oof(SkLR(max_iter=200), X, y, n_splits=3)
    assert oof.shape == (n, 3)
    # rows scored by a 2-class fold have zero in the missing class column
    assert np.all(oof.sum(axis=1) <= 1.0 + 1e-9)
    assert hasattr(fitted, "classes_")


def test_run_tasks_wraps_failures_in_world1():
    from skdist_amd.parallel.cluster import Cluster, TaskFailedError

    sc = Cluster(device="cpu")
    assert not sc.distributed

    def boom(task):
        if task == 1:
            raise ValueError("kaboom")
        return task * 10

    with pytest.raises(TaskFailedError, match="kaboom"):
        sc.run_tasks(boom, [0, 1, 2])
    assert sc.run_tasks(lambda t: t + 1, [1, 2, 3]) == [2, 3, 4]


def test_fit_param_slicing_all_meta_estimators():
    """Round-2 audit fix: sample-aligned fit params (sample_weight) are
    fold-sliced in EVERY per-task worker (search / multimodel /
    eliminator) — previously a shape crash."""
    from sklearn.linear_model import LogisticRegression as SkLR

    from skdist_amd.distribute.eliminate import DistFeatureEliminator
    from skdist_amd.distribute.search import (
        DistGridSearchCV,
        DistMultiModelSearch,
    )

    rng = np.random.default_rng(0)
    X = rng.standard_normal((240, 5))
    y = (X[:, 0] > 0).astype(int)
    w = rng.random(240)
    gs = DistGridSearchCV(SkLR(max_iter=100), {"C": [1.0]}, cv=3)
    gs.fit(X, y, sample_weight=w)
    assert gs.best_score_ > 0.8
    mm = DistMultiModelSearch(
        [("lr", SkLR(max_iter=100), {"C": [0.1, 1.0]})], n=2, cv=3)
    mm.fit(X, y, sample_weight=w)
    assert mm.best_score_ > 0.8
    fe = DistFeatureEliminator(
        SkLR(max_iter=100), min_features_to_select=3, cv=3)
    fe.fit(X, y, sample_weight=w)
    assert fe.best_score_ > 0.8
    # non-sample-aligned fit params pass through untouched
    from skdist_amd.distribute.search import _slice_fit_params

    fp = {"eval_set": [(X[:10], y[:10])], "classes": [0, 1],
          "sample_weight": w}
    out = _slice_fit_params(fp, np.arange(100), 240)
    assert out["eval_set"] is fp["eval_set"]
    assert out["classes"] is fp["classes"]
    assert len(out["sample_weight"]) == 100


def test_nan_inputs_raise_like_sklearn():
    """Round-2 audit fix: NaN/inf inputs raise instead of silently
    training NaN models (device-side finite check in every dataset
    builder)."""
    import scipy.sparse as sp

    from skdist_amd.models import (
        HistGradientBoostingClassifier,
        LogisticRegression,
        Ridge,
    )

    rng = np.random.default_rng(0)
    X = rng.standard_normal((100, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(int)
    Xn = X.copy()
    Xn[3, 2] = np.nan
    with pytest.raises(ValueError, match="NaN or infinity"):
        LogisticRegression(epochs=2).fit(Xn, y)
    with pytest.raises(ValueError, match="NaN or infinity"):
        Ridge(epochs=2).fit(
            X, np.where(np.arange(100) == 5, np.inf, y.astype(float)))
    with pytest.raises(ValueError, match="NaN or infinity"):
        HistGradientBoostingClassifier(n_estimators=2).fit(Xn, y)
    with pytest.raises(ValueError, match="NaN or infinity"):
        import os

        os.environ["SKDIST_AMD_FORCE_SPARSE"] = "1"
        try:
            LogisticRegression(epochs=2, momentum=0).fit(
                sp.csr_matrix(Xn), y)
        finally:
            os.environ.pop("SKDIST_AMD_FORCE_SPARSE")


def test_predict_before_fit_raises_notfitted():
    """Round-2 audit fix: unfitted predict raises sklearn's
    NotFittedError (was a bare AttributeError, invisible to
    ``except NotFittedError``)."""
    from sklearn.exceptions import NotFittedError

    from skdist_amd.distribute.eliminate import DistFeatureEliminator
    from skdist_amd.distribute.ensemble import DistRandomForestClassifier
    from skdist_amd.distribute.multiclass import (
        DistOneVsOneClassifier,
        DistOneVsRestClassifier,
    )
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import (
        HistGradientBoostingClassifier,
        LogisticRegression,
    )

    X = np.zeros((4, 3), dtype=np.float32)
    for m in (LogisticRegression(), HistGradientBoostingClassifier(),
              DistRandomForestClassifier(n_estimators=2),
              DistOneVsRestClassifier(LogisticRegression()),
              DistOneVsOneClassifier(LogisticRegression()),
              DistFeatureEliminator(LogisticRegression()),
              DistGridSearchCV(LogisticRegression(), {"C": [1.0]})):
        with pytest.raises(NotFittedError):
            m.predict(X)
