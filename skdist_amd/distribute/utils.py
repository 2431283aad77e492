"""
CV scoring / splitting utilities (reference: skdist/distribute/utils.py).

The reference vendored sklearn-internal helpers to survive old sklearn
versions (``_check_multimetric_scoring`` utils.py:75-143, ``_safe_split``
utils.py:171-209, ``_score`` utils.py:45-72).  We implement the same
behavior against the modern public sklearn API instead of copying privates.
"""

import numbers

import numpy as np
import scipy.sparse as sp
from sklearn.metrics import check_scoring, get_scorer


def _aggregate_score_dicts(scores):
    """[{'a': 1}, {'a': 2}] -> {'a': array([1, 2])} (utils.py:13-15)."""
    if not scores:
        return {}
    return {k: np.asarray([s[k] for s in scores]) for k in scores[0]}


def _check_multimetric_scoring(estimator, scoring=None):
    """Normalize ``scoring`` into (dict_of_scorers, is_multimetric)."""
    if scoring is None or isinstance(scoring, str) or callable(scoring):
        return {"score": check_scoring(estimator, scoring=scoring)}, False
    if isinstance(scoring, (list, tuple, set)):
        names = list(scoring)
        if len(set(names)) != len(names):
            raise ValueError(f"duplicate scorer names in {names!r}")
        if not names:
            raise ValueError("empty scoring list")
        return {str(n): get_scorer(n) for n in names}, True
    if isinstance(scoring, dict):
        out = {}
        for name, sc in scoring.items():
            out[str(name)] = sc if callable(sc) else get_scorer(sc)
        if not out:
            raise ValueError("empty scoring dict")
        return out, True
    raise ValueError(f"invalid scoring: {scoring!r}")


def _scorer_accepts_sample_weight(scorer):
    """sklearn 1.7 semantics: a Scorer advertises acceptance via
    _accept_sample_weight(); bare callables by signature."""
    import inspect

    if hasattr(scorer, "_accept_sample_weight"):
        try:
            return bool(scorer._accept_sample_weight())
        except Exception:
            return False
    try:
        return "sample_weight" in inspect.signature(scorer).parameters
    except (TypeError, ValueError):
        return False


def _score(estimator, X_test, y_test, scorers, sample_weight=None):
    """Score a fitted estimator on a test fold with one or more scorers.

    Returns {name: float}.  Mirrors reference utils.py:45-72 including the
    "score must be a number" check.  ``sample_weight`` (already sliced to
    the fold) forwards to each scorer that accepts it — sklearn's
    routing-disabled BaseSearchCV behavior.
    """
    out = {}
    for name, scorer in scorers.items():
        kw = {}
        if sample_weight is not None and _scorer_accepts_sample_weight(
                scorer):
            kw["sample_weight"] = sample_weight
        if y_test is None:
            s = scorer(estimator, X_test, **kw)
        else:
            s = scorer(estimator, X_test, y_test, **kw)
        if hasattr(s, "item"):
            s = s.item()
        if not isinstance(s, numbers.Number):
            raise ValueError(
                f"scoring must return a number, got {s!r} ({type(s)}) "
                f"instead. (scorer={name})"
            )
        out[name] = float(s)
    return out


def _num_samples(x):
    """Number of samples in array-like x (reference utils.py:146-168)."""
    if hasattr(x, "fit") and callable(x.fit):
        raise TypeError(f"Expected sequence or array-like, got estimator {x}")
    if hasattr(x, "shape"):
        if len(x.shape) == 0:
            raise TypeError(f"Singleton array {x!r} cannot be considered a valid collection.")
        return x.shape[0]
    if not hasattr(x, "__len__"):
        raise TypeError(f"Expected sequence or array-like, got {type(x)}")
    return len(x)


def _safe_indexing(X, indices):
    """Row-subset X for numpy / scipy sparse / pandas / python lists
    (reference validation.py:146-264 condensed to the cases we support).
    """
    if X is None:
        return None
    if hasattr(X, "iloc"):
        return X.iloc[indices]
    if sp.issparse(X):
        return X[indices]
    if hasattr(X, "shape"):
        return X[indices]
    return [X[i] for i in indices]


def _safe_split(estimator, X, y, indices, train_indices=None):
    """Create a train/test subset honoring precomputed kernels
    (reference utils.py:171-209).
    """
    if getattr(estimator, "kernel", None) == "precomputed" or getattr(
        estimator, "_pairwise", False
    ):
        if not hasattr(X, "shape"):
            raise ValueError(
                "Precomputed kernels or affinity matrices have to be passed "
                "as arrays or sparse matrices."
            )
        if X.shape[0] != X.shape[1]:
            raise ValueError("X should be a square kernel matrix")
        if train_indices is None:
            X_subset = X[np.ix_(indices, indices)]
        else:
            X_subset = X[np.ix_(indices, train_indices)]
    else:
        X_subset = _safe_indexing(X, indices)
    y_subset = _safe_indexing(y, indices) if y is not None else None
    return X_subset, y_subset


def _dict_slice_remove(d, remove_keys):
    """Return a copy of d without the given keys (reference utils.py:212-223)."""
    return {k: v for k, v in d.items() if k not in remove_keys}
