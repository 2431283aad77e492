"""
Argument validation helpers (reference: skdist/distribute/validation.py).
"""

import numpy as np
from sklearn.utils.validation import check_is_fitted as _sk_check_is_fitted


def _check_estimator(obj, verbose=False):
    """Announce which backend a fit will use (reference validation.py:14-20)."""
    if verbose:
        sc = getattr(obj, "sc", None)
        if sc is None:
            print("Using local backend (sc=None)")
        else:
            print(f"Using GPU cluster backend: {sc}")


def _check_is_fitted(estimator, attributes=None):
    """Version-portable fitted check (reference validation.py:23-29)."""
    return _sk_check_is_fitted(estimator, attributes)


def _validate_names(names):
    """Unique, non-conflicting model names (reference validation.py:44-62)."""
    if len(set(names)) != len(names):
        raise ValueError(f"Names provided are not unique: {list(names)!r}")
    invalid = [n for n in names if "__" in str(n)]
    if invalid:
        raise ValueError(f"Estimator names must not contain __: got {invalid!r}")


def _validate_models(models, obj):
    """Validate DistMultiModelSearch's model list
    (reference validation.py:32-96): each entry is
    (name, estimator, param_distributions [, n_iter]).
    """
    if not models:
        raise ValueError("models must be a non-empty list")
    norm = []
    for i, m in enumerate(models):
        m = tuple(m)
        if len(m) == 3:
            name, est, params = m
            n = None
        elif len(m) == 4:
            name, est, params, n = m
        else:
            raise ValueError(
                f"model {i}: expected (name, estimator, params[, n]) tuple, got {m!r}"
            )
        if not hasattr(est, "fit"):
            raise ValueError(f"model {name!r}: estimator has no fit method")
        if not isinstance(params, dict):
            raise ValueError(f"model {name!r}: params must be a dict")
        norm.append((str(name), est, params, n))
    _validate_names([m[0] for m in norm])
    return norm


def _check_n_iter(n_iter, param_distributions):
    """Cap n_iter at the size of a fully-enumerable grid
    (reference validation.py:99-110).
    """
    all_lists = all(
        not hasattr(v, "rvs") for v in param_distributions.values()
    )
    if all_lists:
        size = int(np.prod([len(v) for v in param_distributions.values()]))
        return min(n_iter, size)
    return n_iter


def _require_fitted(est, attr):
    """Raise sklearn's NotFittedError when ``attr`` is absent — our
    re-implemented predict paths previously surfaced a bare
    AttributeError (NotFittedError subclasses it, so ``except
    NotFittedError`` in user code would not fire)."""
    if not hasattr(est, attr):
        from sklearn.exceptions import NotFittedError

        raise NotFittedError(
            f"This {type(est).__name__} instance is not fitted yet. "
            "Call 'fit' before using this estimator."
        )
