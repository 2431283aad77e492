"""
Preprocessing / postprocessing tests (reference mirror:
skdist/tests/test_preprocessing.py + test_postprocessing.py).
"""

import numpy as np
import pandas as pd
import pytest
import scipy.sparse as sp
from sklearn.datasets import load_iris
from sklearn.linear_model import LogisticRegression as SkLogReg

from skdist_amd.postprocessing import SimpleVoter
from skdist_amd.preprocessing import (
    DenseTransformer,
    FeatureCast,
    HashingVectorizerChunked,
    ImputeNull,
    LabelEncoderPipe,
    MultihotEncoder,
    SelectField,
    SelectorMem,
    SparseTransformer,
)


def test_select_field():
    df = pd.DataFrame({"a": [1, 2], "b": [3, 4], "c": [5, 6]})
    assert SelectField(["a", "b"]).fit_transform(df).shape == (2, 2)
    assert SelectField(["a"]).fit_transform(df).shape == (2, 1)
    assert SelectField(["a"], single_dimension=True).fit_transform(
        df
    ).shape == (2,)
    assert SelectField().fit_transform(df).shape == (2, 3)


def test_dense_sparse_roundtrip():
    X = np.eye(3)
    s = SparseTransformer().fit_transform(X)
    assert sp.issparse(s)
    d = DenseTransformer().fit_transform(s)
    assert not sp.issparse(d)
    assert np.allclose(np.asarray(d), X)
    # passthrough cases
    assert sp.issparse(SparseTransformer().fit_transform(s))
    assert not sp.issparse(DenseTransformer().fit_transform(X))


def test_feature_cast():
    X = np.array([[1.7, 2.2]])
    out = FeatureCast(int).fit_transform(X)
    assert out.dtype.kind == "i"
    assert FeatureCast().fit_transform(X) is X


def test_impute_null():
    X = np.array([[1.0, np.nan], [None, 2.0]], dtype=object)
    out = ImputeNull(0).fit_transform(X)
    assert out[0][1] == 0 and out[1][0] == 0


def test_label_encoder_pipe():
    out = LabelEncoderPipe().fit_transform(["a", "b", "a"])
    assert out.shape == (3, 1)
    assert out.ravel().tolist() == [0, 1, 0]


def test_selector_mem():
    X, y = load_iris(return_X_y=True)
    sel = SelectorMem(selector="kbest", threshold=2)
    out = sel.fit_transform(X, y)
    assert out.shape == (150, 2)


def test_hashing_vectorizer_chunked():
    docs = ["the quick brown fox", "jumps over the lazy dog"] * 10
    hv = HashingVectorizerChunked(chunksize=7, n_features=128)
    out = hv.transform(docs)
    assert out.shape == (20, 128)
    hv2 = HashingVectorizerChunked(chunksize=1000, n_features=128)
    out2 = hv2.transform(docs)
    assert (out != out2).nnz == 0
    with pytest.raises(ValueError):
        hv.transform("a single string")


def test_multihot_encoder():
    X = [["a", "b"], ["b"], ["a", "c"]]
    enc = MultihotEncoder().fit(X)
    out = enc.transform(X)
    assert out.shape == (3, 3)
    assert out.sum() == 5
    enc_sp = MultihotEncoder(sparse_output=True).fit(X)
    assert sp.issparse(enc_sp.transform(X))


def test_simple_voter_hard_and_soft():
    X, y = load_iris(return_X_y=True)
    e1 = SkLogReg(solver="liblinear", C=0.1).fit(X, y)
    e2 = SkLogReg(solver="liblinear", C=10.0).fit(X, y)
    hard = SimpleVoter([("a", e1), ("b", e2)], classes=e1.classes_,
                       voting="hard")
    preds = hard.fit(X, y).predict(X)
    assert (preds == y).mean() > 0.9
    with pytest.raises(AttributeError):
        hard.predict_proba(X)

    soft = SimpleVoter([("a", e1), ("b", e2)], classes=e1.classes_,
                       voting="soft", weights=[0.3, 0.7])
    p = soft.predict_proba(X)
    assert p.shape == (150, 3)
    assert np.allclose(p.sum(axis=1), 1.0)
    assert (soft.predict(X) == y).mean() > 0.9
    assert "a" in soft.named_estimators
