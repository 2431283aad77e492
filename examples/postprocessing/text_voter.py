"""End-to-end text voting pipeline (reference analog:
examples/postprocessing/simple_voter.py — the reference's biggest quoted
pipeline: word-hash grid 92.2 s + char-hash grid 148.0 s + 1000-tree ERT
2.4 s = 242.6 s total on a 32-core Spark cluster, 20newsgroups
atheism-vs-religion, holdout ROC AUC 0.849/0.845/0.896 voter).

No network for 20newsgroups, so this generates a synthetic two-class
corpus (class-biased token vocabularies, Zipf-ish frequencies) and runs
the same three pipelines on the MI355X engine:

  1. word model: device hash-vectorize (hash_kernels.hip) at two
     vectorizer configs x C-grid LogisticRegression searches on the
     sparse-native solver;
  2. char model: char_wb n-gram hashing, same search;
  3. tree model: count-vectorize -> SelectKBest -> 1000
     DistExtraTreesClassifier trees on the device histogram builder;
  4. SimpleVoter(soft) over the three, scored on a holdout.
"""

import time

import numpy as np
from sklearn.feature_extraction.text import CountVectorizer
from sklearn.feature_selection import SelectKBest, f_classif
from sklearn.metrics import f1_score, roc_auc_score
from sklearn.preprocessing import LabelEncoder

from skdist_amd.distribute.ensemble import DistExtraTreesClassifier
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression
from skdist_amd.postprocessing import SimpleVoter
from skdist_amd.preprocessing import HashingVectorizerChunked


def make_corpus(n, seed=0):
    rng = np.random.default_rng(seed)
    shared = [f"word{i}" for i in range(3000)]
    v0 = [f"zeta{i}" for i in range(800)]
    v1 = [f"theo{i}" for i in range(800)]
    docs, ys = [], []
    for i in range(n):
        c = int(rng.random() < 0.5)
        vocab = v1 if c else v0
        k_bias = rng.integers(2, 9)
        toks = list(rng.choice(shared, size=60)) + list(
            rng.choice(vocab, size=k_bias))
        rng.shuffle(toks)
        docs.append(" ".join(toks))
        ys.append(c)
    return docs, np.array(ys, dtype=np.int64)


class _Vectorized:
    """Fitted (vectorizer, search) pair exposing predict_proba on raw
    docs — what the voter consumes."""

    def __init__(self, vec, model):
        self.vec = vec
        self.model = model
        self.classes_ = model.best_estimator_.classes_

    def predict_proba(self, docs):
        return self.model.predict_proba(self.vec.transform(docs))

    def predict(self, docs):
        return self.model.predict(self.vec.transform(docs))

    def fit(self, X=None, y=None):
        return self


def search_over_vectorizers(docs, y, configs, grid, sc_factory, cv=3):
    """The reference grids vectorizer params inside one search; the
    MI355X-first layout vectorizes ONCE per config (device hashing
    kernel — features never leave HBM-bound CSR) and batches the whole
    C-grid per config, then compares configs by CV score."""
    best = None
    for kw in configs:
        vec = HashingVectorizerChunked(chunksize=None, **kw)
        Xv = vec.transform(docs)
        gs = DistGridSearchCV(
            LogisticRegression(epochs=10, momentum=0.0, random_state=0),
            grid, cv=cv, scoring="roc_auc", sc=sc_factory())
        gs.fit(Xv, y)
        if best is None or gs.best_score_ > best[0].best_score_:
            best = (gs, vec)
    return _Vectorized(best[1], best[0])


def main():
    import torch

    on_gpu = torch.cuda.is_available()

    def sc_factory():
        from skdist_amd import Cluster

        return Cluster() if on_gpu else None

    n = 20_000 if on_gpu else 800
    docs, y = make_corpus(n + n // 4)
    docs_tr, y_tr = docs[:n], y[:n]
    docs_te, y_te = docs[n:], y[n:]
    grid = {"C": list(np.logspace(-2, 2, 8 if on_gpu else 2))}

    t0 = time.time()
    word = search_over_vectorizers(
        docs_tr, y_tr,
        ([dict(n_features=2 ** 18, ngram_range=(1, 1)),
          dict(n_features=2 ** 20, ngram_range=(1, 2))] if on_gpu
         else [dict(n_features=2 ** 18, ngram_range=(1, 1))]),
        grid, sc_factory)
    t_word = time.time() - t0

    t0 = time.time()
    char = search_over_vectorizers(
        docs_tr, y_tr,
        [dict(analyzer="char_wb",
              n_features=2 ** 20 if on_gpu else 2 ** 18,
              ngram_range=(2, 4) if on_gpu else (2, 3))],
        grid, sc_factory)
    t_char = time.time() - t0

    t0 = time.time()
    cv_vec = CountVectorizer(max_features=30_000)
    Xc = cv_vec.fit_transform(docs_tr)
    kb = SelectKBest(f_classif, k=min(1000, Xc.shape[1] - 1)).fit(
        Xc, y_tr)
    Xk = np.asarray(kb.transform(Xc).todense(), dtype=np.float32)
    ert = DistExtraTreesClassifier(
        n_estimators=1000 if on_gpu else 50, max_depth=20,
        random_state=0, sc=sc_factory()).fit(Xk, y_tr)

    class _TreePipe:
        classes_ = ert.classes_

        def predict_proba(self, docs):
            return ert.predict_proba(np.asarray(
                kb.transform(cv_vec.transform(docs)).todense(),
                dtype=np.float32))

        def fit(self, X=None, y=None):
            return self

    tree = _TreePipe()
    t_tree = time.time() - t0
    total = t_word + t_char + t_tree

    le = LabelEncoder().fit(y_tr)
    voter = SimpleVoter(
        [("word", word), ("char", char), ("tree", tree)],
        classes=le.classes_, voting="soft").fit(None, None)

    print(f"Word Model Fit Time: {t_word:.2f}s")
    print(f"Char Model Fit Time: {t_char:.2f}s")
    print(f"Tree Model Fit Time: {t_tree:.2f}s")
    print(f"Total Fit Time: {total:.2f}s   "
          "(reference pipeline: 242.6 s on a 32-core Spark cluster)")
    for name, m in [("Word", word), ("Char", char), ("Tree", tree),
                    ("Voter", voter)]:
        p = m.predict_proba(docs_te)[:, 1]
        auc = roc_auc_score(y_te, p)
        f1 = f1_score(y_te, (p > 0.5).astype(int))
        print(f"-- {name} --  ROC AUC {auc:.4f}  F1 {f1:.4f}")


if __name__ == "__main__":
    main()
