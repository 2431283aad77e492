"""Compact committed versions of the round-2 hardening sweeps (the
full-size one-off runs are recorded in SURVEY.md §9: AUC fuzz 200/200,
sparse-vs-dense 30/30, weighted forests 12/12, GBT 8/8, vectorizers
50/50, OvR/OvO agreement 12/12, determinism 15/15, guard fuzz 25/25)."""

import os
import warnings

import numpy as np
import pytest
import scipy.sparse as sp
import torch
from sklearn.metrics import roc_auc_score


def test_auc_fuzz_compact():
    from skdist_amd.models._sgd import _MetricState

    rng = np.random.default_rng(0)
    for trial in range(20):
        m = int(rng.integers(30, 800))
        nm = int(rng.integers(1, 6))
        q = rng.choice([1, 2, 100])
        Z = torch.as_tensor(
            np.round(rng.standard_normal((m, nm)) * q) / q,
            dtype=torch.float32)
        y = torch.as_tensor(
            (rng.random(m) < rng.uniform(0.1, 0.9)).astype(np.float32))
        if y.sum() in (0, m):
            continue

        class S:
            pass

        spec = S()
        spec.col_class = torch.ones(nm, dtype=torch.int32)
        fid = torch.zeros(m, dtype=torch.int32)
        mf = torch.zeros(nm, dtype=torch.int32)
        st = _MetricState("roc_auc", nm, 2, torch.device("cpu"))
        cut = m // 2
        st.update(Z[:cut], y[:cut], fid[:cut], spec, mf, 2)
        st.update(Z[cut:], y[cut:], fid[cut:], spec, mf, 2)
        ours = st.finalize()
        for c in range(nm):
            expect = roc_auc_score(y.numpy(), Z[:, c].numpy())
            assert abs(ours[c] - expect) < 1e-10


def test_sparse_vs_dense_fuzz_compact(monkeypatch):
    from sklearn.model_selection import KFold

    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LinearSVC, LogisticRegression

    rng = np.random.default_rng(1)
    for trial in range(4):
        n = int(rng.integers(500, 1500))
        f = int(rng.integers(30, 120))
        Xd = rng.standard_normal((n, f)).astype(np.float32)
        Xd[np.abs(Xd) < rng.uniform(0.6, 1.2)] = 0
        w = rng.standard_normal(f) * (rng.random(f) < 0.3)
        y = ((Xd @ w) > 0).astype(int)
        X = sp.csr_matrix(Xd)
        cv = KFold(int(rng.integers(2, 4)))
        Cs = list(10.0 ** rng.uniform(-1, 1, size=2))
        cls = [LogisticRegression, LinearSVC][trial % 2]
        kw = dict(epochs=6, momentum=0.0, random_state=0,
                  adaptive=False)
        monkeypatch.setenv("SKDIST_AMD_FORCE_SPARSE", "1")
        g_sp = DistGridSearchCV(cls(**kw), {"C": Cs}, cv=cv).fit(X, y)
        monkeypatch.setenv("SKDIST_AMD_FORCE_SPARSE", "0")
        g_de = DistGridSearchCV(
            cls(standardize=False, **kw), {"C": Cs}, cv=cv).fit(Xd, y)
        np.testing.assert_allclose(
            g_sp.cv_results_["mean_test_score"],
            g_de.cv_results_["mean_test_score"], atol=8e-3)


def test_weighted_forest_quality_compact():
    from sklearn.ensemble import RandomForestClassifier

    from skdist_amd.models.forest import BinnedDataset, ForestBuilder

    rng = np.random.default_rng(2)
    for trial in range(3):
        n, f = 3000, 10
        X = rng.standard_normal((n, f)).astype(np.float32)
        y = ((X @ rng.standard_normal(f)) > 0).astype(np.int64)
        sw = rng.random(n) * 2 + 0.2
        ds = BinnedDataset(X, y, "cpu", is_cls=True)
        b = ForestBuilder(ds, "gini", max_depth=8, min_samples_split=4,
                          min_samples_leaf=2, max_features="sqrt",
                          extra_mode=False, bootstrap=True)
        trees = b.build(list(range(8)), sample_weight=sw)
        proba = np.mean([t.predict_proba(X) for t in trees], axis=0)
        acc = ((proba[:, 1] > 0.5).astype(int) == y).mean()
        sk = RandomForestClassifier(
            n_estimators=8, max_depth=8, random_state=0, n_jobs=2
        ).fit(X, y, sample_weight=sw)
        assert acc > (sk.predict(X) == y).mean() - 0.05
