"""
Encoderizer — feature union with per-column encoder inference
(reference: skdist/distribute/encoder.py).

A FeatureUnion-style transformer that accepts pandas / dict / numpy / list
input, infers an encoder pipeline per column (or takes an explicit
``config`` / ``transformer_list``), and fits the per-column transformers
as independent tasks over the scheduler (parallel axis P7,
reference encoder.py:137-153).  Implemented standalone — not on top of
sklearn's FeatureUnion internals — so it is sklearn-version-proof.
"""

import ast
from copy import copy

import numpy as np
import pandas as pd
import scipy.sparse as sparse
from sklearn.base import BaseEstimator, TransformerMixin

from ..parallel.local import run_local_tasks
from .base import _strip_sc

__all__ = ["Encoderizer", "EncoderizerExtractor"]

_CATEGORICAL_UNIQUE_RATIO = 0.10  # reference encoder.py:368-371


class Encoderizer(TransformerMixin, BaseEstimator):
    """Flexible-input feature encoder (reference encoder.py:33-387).

    Args:
        transformer_list: explicit [(name, transformer)] list (skips
            inference).
        transformer_weights: {name: multiplicative weight}.
        n_jobs: joblib jobs for the local path.
        size: 'small' | 'medium' | 'large' — default-encoder tier.
        config: {column: encoder kind} to force encoder types.
        col_names: column names for bare numpy/list input.
        sc: None (local) or a Cluster.
    """

    def __init__(self, transformer_list=None, transformer_weights=None,
                 n_jobs=1, size="small", config=None, col_names=None,
                 sc=None, partitions="auto"):
        self.transformer_list = transformer_list
        self.transformer_weights = transformer_weights
        self.n_jobs = n_jobs
        self.size = size
        self.config = config
        self.col_names = col_names
        self.sc = sc
        self.partitions = partitions

    # ------------------------------------------------------------------ #
    @property
    def step_names(self):
        return [name for name, _ in self.transformer_list]

    def fit(self, X, y=None):
        X = self._process_input(X)
        if self.transformer_list is None:
            self.transformer_list = self._infer_transformers(X)
        self.transformer_list = list(self.transformer_list)
        if not self.transformer_list:
            raise ValueError("no transformers to fit")

        def task_fn(task):
            idx, trans = task
            return idx, trans.fit(X, y)

        tasks = list(
            (i, t) for i, (_, t) in enumerate(self.transformer_list)
        )
        sc = self.sc
        if sc is None:
            results = run_local_tasks(task_fn, tasks, n_jobs=self.n_jobs)
        else:
            results = sc.run_tasks(task_fn, tasks)
        results.sort(key=lambda r: r[0])
        self.transformer_list = [
            (self.transformer_list[i][0], fitted) for i, fitted in results
        ]
        self._feature_indices(X)
        _strip_sc(self)
        return self

    def transform(self, X):
        X = self._process_input(X, fit=False)
        outs = []
        for name, trans in self.transformer_list:
            out = trans.transform(X)
            w = (self.transformer_weights or {}).get(name)
            if w is not None:
                out = out * w
            outs.append(out)
        if not outs:
            return np.zeros((X.shape[0], 0))
        if any(sparse.issparse(o) for o in outs):
            return sparse.hstack(outs).tocsr()
        return np.hstack(outs)

    def fit_transform(self, X, y=None, **fit_params):
        return self.fit(X, y).transform(X)

    # ------------------------------------------------------------------ #
    def extract(self, step_names):
        """Copy of fitted self restricted to the given steps
        (reference encoder.py:88-110)."""
        out = copy(self)
        keep = [i for i, n in enumerate(self.step_names) if n in step_names]
        out.transformer_list = [self.transformer_list[i] for i in keep]
        out.transformer_lengths = [self.transformer_lengths[i] for i in keep]
        return out

    def feature_origin(self, index, mask=None):
        """Step name that produced transformed-feature ``index``
        (reference encoder.py:209-230)."""
        cumulative = np.cumsum(self.transformer_lengths)
        if mask is not None:
            cumulative = np.array([mask[x - 1] for x in cumulative])
        return self.step_names[int(np.argmax(cumulative > index))]

    # ------------------------------------------------------------------ #
    def _process_input(self, X, fit=True):
        """pandas / dict / numpy / list → DataFrame
        (reference encoder.py:237-266)."""
        if isinstance(X, pd.DataFrame):
            out = X
        elif isinstance(X, dict):
            out = pd.DataFrame.from_dict(X, orient="columns")
        elif isinstance(X, list) and X and isinstance(X[0], dict):
            # list of record dicts → one column per key
            out = pd.DataFrame.from_records(X)
        elif isinstance(X, (np.ndarray, list)):
            if fit:
                if self.col_names is None:
                    raise ValueError(
                        "Must supply col_names with numpy array input"
                    )
                out = pd.DataFrame(X, columns=self.col_names)
            else:
                out = pd.DataFrame(X, columns=self.fields_)
        else:
            raise ValueError(f"Cannot parse input of type {type(X)}")
        if fit:
            self.fields_ = list(out.columns)
        return out

    def _infer_transformers(self, X):
        from ._defaults import _default_encoders

        registry = _default_encoders[self.size]
        if self.config is not None:
            bad_cols = [c for c in self.config if c not in X.columns]
            if bad_cols:
                raise ValueError(
                    f"config names columns not in the input: {bad_cols} "
                    f"(available: {list(X.columns)})"
                )
            bad_kinds = {
                k for k in self.config.values() if k not in registry
            }
            if bad_kinds:
                raise ValueError(
                    f"unknown encoder kind(s) {sorted(bad_kinds)}; "
                    f"choose from {sorted(registry)}"
                )
            groups = [registry[kind](col) for col, kind in
                      self.config.items()]
        else:
            groups = [
                self._infer_column(c, X[c], registry) for c in X.columns
            ]
        return [
            step for group in groups if group is not None for step in group
        ]

    @staticmethod
    def _first_non_null(col):
        vals = col.values
        for v in vals:
            if v is not None:
                return v
        return None

    @classmethod
    def _container_kind(cls, col, col_name):
        """dict/list/tuple detection with the reference's stringified-
        container guard (reference encoder.py:281-342)."""
        v = cls._first_non_null(col)
        if isinstance(v, str):
            try:
                ast.literal_eval(v)
            except (ValueError, SyntaxError):
                return None
            raise ValueError(
                f"Convert this column to its container type before "
                f"fitting: {col_name}"
            )
        if isinstance(v, dict):
            return "dict"
        if isinstance(v, (list, tuple)):
            return "multihotencoder"
        return None

    def _infer_column(self, col_name, col, registry,
                      thresh=_CATEGORICAL_UNIQUE_RATIO):
        """Pick an encoder kind for one column
        (reference encoder.py:344-377)."""
        if col.isnull().all():
            return None
        kind = self._container_kind(col, col_name)
        if kind is not None:
            return registry[kind](col_name)
        try:
            np.mean(col)
            is_numeric = True
        except Exception:
            is_numeric = False
        pct_unique = col.nunique() / float(len(col))
        is_categorical = pct_unique < thresh
        if not is_numeric and not is_categorical:
            return registry["string_vectorizer"](col_name)
        if is_numeric and not is_categorical:
            return registry["numeric"](col_name)
        return registry["onehotencoder"](col_name)

    def _feature_indices(self, X):
        widths = []
        for _, trans in self.transformer_list:
            head = trans.transform(X.head(1))
            widths.append(
                len(head[0]) if isinstance(head, list) else head.shape[1]
            )
        self.transformer_lengths = widths


class EncoderizerExtractor(TransformerMixin, BaseEstimator):
    """Pipeline-safe slice of a fitted Encoderizer
    (reference encoder.py:390-411)."""

    def __init__(self, encoderizer, step_names):
        self.encoderizer = encoderizer.extract(step_names)

    def fit(self, X, y=None):
        return self

    def transform(self, X):
        return self.encoderizer.transform(X)
