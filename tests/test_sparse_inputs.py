"""Scipy-sparse input through the native estimators and meta-estimators.

The reference's headline multiclass/text workloads feed CSR matrices
(hashed text features) into OvR/search (reference skdist/distribute/
multiclass.py binarizes y via CSC; encoder output is sparse hstack,
encoder.py:182-185).  Our batched solver is dense-MFMA by design, so
sparse X densifies into the device dataset (288 GB HBM — SURVEY.md §2.3)
behind a size guard; host inference stays sparse-aware.
"""

import pickle

import numpy as np
import pytest
import scipy.sparse as sp

from skdist_amd.distribute.multiclass import (
    DistOneVsOneClassifier,
    DistOneVsRestClassifier,
)
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LinearSVC, LogisticRegression, Ridge


@pytest.fixture(scope="module")
def sparse_xy():
    rng = np.random.default_rng(0)
    Xd = rng.standard_normal((300, 12)).astype(np.float32)
    Xd[Xd < 0.5] = 0.0  # ~70% zeros
    y = (Xd[:, 0] + Xd[:, 1] > 0.8).astype(np.int64)
    y3 = Xd[:, :3].argmax(axis=1).astype(np.int64)
    return Xd, sp.csr_matrix(Xd), y, y3


def test_native_fit_sparse_equals_dense(sparse_xy):
    Xd, Xs, y, _ = sparse_xy
    m_s = LogisticRegression(epochs=10, random_state=0).fit(Xs, y)
    m_d = LogisticRegression(epochs=10, random_state=0).fit(Xd, y)
    assert np.allclose(m_s.coef_, m_d.coef_)
    assert np.allclose(m_s.intercept_, m_d.intercept_)
    # sparse and dense inference agree
    assert np.allclose(m_s.predict_proba(Xs), m_d.predict_proba(Xd))
    assert (m_s.predict(Xs) == m_d.predict(Xd)).all()


@pytest.mark.parametrize("cls", [LinearSVC, LogisticRegression])
def test_native_multiclass_sparse(sparse_xy, cls):
    _, Xs, _, y3 = sparse_xy
    m = cls(epochs=10, random_state=0).fit(Xs, y3)
    assert (m.predict(Xs) == y3).mean() > 0.9


def test_ridge_sparse(sparse_xy):
    Xd, Xs, _, _ = sparse_xy
    t = (Xd @ np.arange(12)).astype(np.float64)
    r = Ridge(epochs=10, random_state=0).fit(Xs, t)
    assert np.mean((r.predict(Xs) - t) ** 2) < 0.05 * t.var()


def test_search_batched_sparse(sparse_xy):
    Xd, Xs, y, _ = sparse_xy
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3,
    )
    gs.fit(Xs, y)
    gd = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3,
    )
    gd.fit(Xd, y)
    assert np.allclose(
        gs.cv_results_["mean_test_score"], gd.cv_results_["mean_test_score"]
    )
    assert (gs.predict(Xs[:20]) == gd.predict(Xd[:20])).all()


def test_ovr_ovo_sparse(sparse_xy):
    _, Xs, _, y3 = sparse_xy
    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=10, random_state=0), norm="l1"
    ).fit(Xs, y3)
    assert (ovr.predict(Xs) == y3).mean() > 0.9
    proba = ovr.predict_proba(Xs)
    assert np.allclose(proba.sum(axis=1), 1.0)
    ovo = DistOneVsOneClassifier(
        LogisticRegression(epochs=10, random_state=0)
    ).fit(Xs, y3)
    assert (ovo.predict(Xs) == y3).mean() > 0.9
    # fitted objects stay pickle-clean with sparse inputs
    blob = pickle.loads(pickle.dumps(ovr))
    assert (blob.predict(Xs) == ovr.predict(Xs)).all()


def test_wide_sparse_routes_to_sparse_native_path():
    """Round 1 raised a densify-size ValueError here; round 2's sparse-
    native solver (models/_sparse_sgd.py) fits this shape directly."""
    rng = np.random.default_rng(0)
    rows = np.repeat(np.arange(10_000), 5)
    cols = rng.integers(0, 3_000_000, size=len(rows))
    vals = rng.standard_normal(len(rows)).astype(np.float32)
    big = sp.csr_matrix((vals, (rows, cols)), shape=(10_000, 3_000_000))
    y = (np.asarray(big[:, :100_000].sum(axis=1)).ravel() > 0).astype(int)
    m = LogisticRegression(epochs=2, momentum=0.0).fit(big, y)
    assert m.coef_.shape == (1, 3_000_000)
    assert (m.predict(big) == y).mean() > 0.6


def test_text_pipeline_sparse_end_to_end():
    """HashingVectorizer CSR output → native search (reference text flow,
    e.g. examples/encoder/basic_usage.py)."""
    from sklearn.feature_extraction.text import HashingVectorizer

    docs = [f"alpha beta topic{i % 2} word{i % 7}" for i in range(240)]
    y = np.array([i % 2 for i in range(240)])
    Xs = HashingVectorizer(n_features=256).transform(docs)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=15, random_state=0),
        {"C": [0.1, 1.0, 10.0]}, cv=3,
    )
    gs.fit(Xs, y)
    assert gs.best_score_ > 0.95
    assert (gs.predict(Xs) == y).mean() > 0.95


def test_encoderizer_to_native_ovr_text_flow():
    """The reference's flagship text workflow: Encoderizer sparse text
    features -> native OvR (densified into the device dataset), with a
    pickled (encoder, model) pair predicting end-to-end."""
    import pandas as pd

    from skdist_amd.distribute.encoder import Encoderizer
    from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
    from skdist_amd.models import LinearSVC

    rng = np.random.default_rng(0)
    topics = ["sports ball game team win",
              "market stock trade price fund",
              "code python bug compile test"]
    docs, labels = [], []
    for i in range(600):
        k = i % 3
        docs.append(" ".join(
            rng.choice(topics[k].split(), size=6)) + f" common{i % 5}")
        labels.append(["sport", "finance", "tech"][k])
    df = pd.DataFrame({"text": docs, "num": rng.standard_normal(600)})
    y = np.array(labels)

    enc = Encoderizer(size="medium")
    T = enc.fit_transform(df)
    assert sp.issparse(T)
    ovr = DistOneVsRestClassifier(
        LinearSVC(epochs=12, random_state=0)
    ).fit(T, y)
    assert (ovr.predict(enc.transform(df)) == y).mean() > 0.98
    blob = pickle.loads(pickle.dumps({"enc": enc, "ovr": ovr}))
    out = blob["ovr"].predict(blob["enc"].transform(df.iloc[:6]))
    assert set(out) <= {"sport", "finance", "tech"}
