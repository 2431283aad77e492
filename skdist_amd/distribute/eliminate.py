"""
Distributed feature elimination (reference: skdist/distribute/eliminate.py).

Approximate RFECV: rank features ONCE from an initial full fit, build a
ladder of leave-k-out feature subsets, then score every (subset × fold)
combination as an independent task over the scheduler
(reference eliminate.py:111-239).  Results are keyed by task id — not by
the reference's float array-equality matching (eliminate.py:211-219).
"""

from itertools import product

import numpy as np
from sklearn.base import BaseEstimator, ClassifierMixin, is_classifier
from sklearn.metrics import check_scoring
from sklearn.model_selection import check_cv
from sklearn.utils import safe_sqr
from sklearn.utils.validation import check_X_y

from ..parallel.local import run_local_tasks
from .base import _clone
from .utils import _safe_split
from .validation import _check_estimator


def _drop_cols(X, drop):
    """X without the given feature columns (reference eliminate.py:23-27)."""
    if len(drop) == 0:
        return X
    keep = np.delete(np.arange(X.shape[1]), drop)
    return X[:, keep]


class DistFeatureEliminator(ClassifierMixin, BaseEstimator):
    """Distributed approximate RFECV (reference eliminate.py:47-284)."""

    def __init__(self, estimator, sc=None, partitions="auto",
                 min_features_to_select=None, step=1, cv=5, scoring=None,
                 verbose=False, n_jobs=None, pre_dispatch=None, mask=True):
        self.estimator = estimator
        self.sc = sc
        self.partitions = partitions
        self.min_features_to_select = min_features_to_select
        self.step = step
        self.cv = cv
        self.scoring = scoring
        self.verbose = verbose
        self.n_jobs = n_jobs
        self.pre_dispatch = pre_dispatch
        self.mask = mask

    def fit(self, X, y=None, groups=None, **fit_params):
        _check_estimator(self, verbose=bool(self.verbose))
        sc = self.sc
        if sc is not None and getattr(sc, "distributed", False):
            X, y, groups = sc.sync_host_data(X, y, groups)
        X, y = check_X_y(X, y, accept_sparse="csr", ensure_min_features=2)
        cv = check_cv(self.cv, y, classifier=is_classifier(self.estimator))
        scorer = check_scoring(self.estimator, scoring=self.scoring)

        n_features = X.shape[1]
        min_keep = (
            n_features // 2 if self.min_features_to_select is None
            else self.min_features_to_select
        )
        if 0.0 < self.step < 1.0:
            step = int(max(1, self.step * n_features))
        else:
            step = int(self.step)
        if step <= 0:
            raise ValueError("Step must be >0")

        # initial full fit ranks the features (reference :141-157)
        initial = _clone(self.estimator)
        if hasattr(initial, "sc"):
            initial.sc = None
        initial.fit(X, y, **fit_params)
        coefs = getattr(initial, "coef_", None)
        if coefs is None:
            coefs = getattr(initial, "feature_importances_", None)
        if coefs is None:
            raise RuntimeError(
                'The estimator exposes neither "coef_" nor '
                '"feature_importances_"'
            )
        coefs = np.asarray(coefs)
        if coefs.ndim > 1:
            ranks = np.argsort(safe_sqr(coefs).sum(axis=0))
        else:
            ranks = np.argsort(safe_sqr(coefs))
        ranks = np.ravel(ranks)[: (n_features - min_keep)]

        removals = [np.array([], dtype=int)]
        removed = 0
        while removed < (n_features - min_keep):
            removed += step
            removals.append(ranks[:removed])

        cv_splits = list(cv.split(X, y, groups))

        # batched device path: ONE masked multi-column solve scores every
        # (subset × fold) — replaces the reference's per-subset Spark
        # tasks (eliminate.py:191-210)
        refit_fn = None
        scores_mat = None
        if (
            sc is not None
            and hasattr(self.estimator, "batched_eliminate")
            and not fit_params
        ):
            from ..models.linear import FallbackToGeneric

            try:
                scores_mat, refit_fn = self.estimator.batched_eliminate(
                    X, y, removals, cv_splits, self.scoring, sc)
            except FallbackToGeneric:
                scores_mat = None

        if scores_mat is not None:
            self.scores_ = [float(m) for m in scores_mat.mean(axis=1)]
        else:
            tasks = [
                (tid, si, split)
                for tid, (si, split) in enumerate(
                    product(range(len(removals)), cv_splits)
                )
            ]

            def task_fn(task):
                tid, si, (train, test) = task
                est = _clone(self.estimator)
                if hasattr(est, "sc"):
                    est.sc = None
                Xs = _drop_cols(X, removals[si])
                X_tr, y_tr = _safe_split(est, Xs, y, train)
                X_te, y_te = _safe_split(est, Xs, y, test, train)
                from .search import _slice_fit_params
                from .utils import _num_samples

                est.fit(X_tr, y_tr, **_slice_fit_params(
                    fit_params, train, _num_samples(Xs)))
                return tid, si, float(scorer(est, X_te, y_te))

            if sc is None:
                results = run_local_tasks(
                    task_fn, tasks, n_jobs=self.n_jobs,
                    pre_dispatch=self.pre_dispatch or "2*n_jobs",
                )
            else:
                results = sc.run_tasks(task_fn, tasks)

            per_set = [[] for _ in removals]
            for tid, si, score in sorted(results, key=lambda r: r[0]):
                per_set[si].append(score)
            self.scores_ = [float(np.mean(s)) for s in per_set]

        # exact ties break toward the SMALLEST feature set (the ladder is
        # ordered by increasing removal) — parsimony over argmax-first
        scores_arr = np.asarray(self.scores_)
        best = int(len(scores_arr) - 1 - np.argmax(scores_arr[::-1]))
        self.best_score_ = self.scores_[best]
        if len(removals[best]) > 0:
            self.best_features_ = np.delete(
                np.arange(n_features), removals[best]
            )
        else:
            self.best_features_ = np.arange(n_features)
        best_est = refit_fn(best) if refit_fn is not None else None
        if best_est is not None:
            # batched path: the subset's full-data model trained alongside
            # the CV columns — refit is a column extraction
            self.best_estimator_ = best_est
        else:
            self.best_estimator_ = _clone(self.estimator)
            if hasattr(self.best_estimator_, "sc"):
                self.best_estimator_.sc = None
            self.best_estimator_.fit(
                X[:, self.best_features_], y, **fit_params)
        self.n_features_ = len(self.best_features_)

        del self.sc
        self.sc = None
        return self

    # ------------------------------------------------------------------ #
    def _mask_X(self, X):
        if self.mask:
            return X[:, self.best_features_]
        return X

    def predict(self, X):
        from .validation import _require_fitted

        _require_fitted(self, "best_estimator_")
        return self.best_estimator_.predict(self._mask_X(X))

    def predict_proba(self, X):
        return self.best_estimator_.predict_proba(self._mask_X(X))

    def predict_log_proba(self, X):
        return self.best_estimator_.predict_log_proba(self._mask_X(X))

    def decision_function(self, X):
        return self.best_estimator_.decision_function(self._mask_X(X))

    def transform(self, X):
        return self._mask_X(X)

    def score(self, X, y):
        scorer = check_scoring(self.best_estimator_, scoring=self.scoring)
        return scorer(self.best_estimator_, self._mask_X(X), y)
