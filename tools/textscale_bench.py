"""Measured text-scale workload: DistGridSearchCV(LogisticRegression)
on a synthetic hashed-text-shaped CSR (n docs x 2^20 features, ~60
nnz/row, l2-normalized rows — the shape sk-dist's HashingVectorizer
pipelines produce, reference _defaults.py:91-198).  There is no network
for a real corpus, so the matrix is synthetic with a planted linear
signal; quality is reported as holdout accuracy.

Runs on GPU (HIP sparse path) when available, CPU eager otherwise.
Prints one JSON line and writes it to gpurun_out/r2_textscale.json.
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import scipy.sparse as sp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_text_csr(n, f, nnz_per_row, seed, signal_feats=20000):
    rng = np.random.default_rng(seed)
    total = n * nnz_per_row
    # Zipf-ish feature popularity: mixture of a hot head and uniform tail
    hot = rng.integers(0, 1 << 14, size=total // 2)
    tail = rng.integers(0, f, size=total - total // 2)
    cols = np.concatenate([hot, tail])
    rng.shuffle(cols)
    # sorted within each row, like real vectorizer CSR output (unsorted
    # indices cost a 0.9 s scipy sort_indices INSIDE the timed fit)
    cols = np.sort(cols.reshape(n, nnz_per_row), axis=1)
    vals = np.full((n, nnz_per_row), 1.0 / np.sqrt(nnz_per_row),
                   dtype=np.float32)
    w = np.zeros(f, dtype=np.float32)
    # signal on the hot head (recurring "tokens" — learnable at any n)
    # plus a thin tail component
    n_hot = min(signal_feats, 1 << 14)
    w[: 1 << 14][rng.permutation(1 << 14)[:n_hot]] = (
        rng.standard_normal(n_hot).astype(np.float32) * 4.0)
    tail_sig = rng.integers(1 << 14, f, size=signal_feats)
    w[tail_sig] = rng.standard_normal(signal_feats).astype(np.float32)
    scores = (w[cols] * vals).sum(axis=1)
    y = (scores + 0.15 * rng.standard_normal(n) > 0).astype(np.int64)
    indptr = np.arange(n + 1, dtype=np.int64) * nnz_per_row
    X = sp.csr_matrix((vals.ravel(), cols.ravel().astype(np.int32),
                       indptr), shape=(n, f))
    return X, y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1_000_000)
    ap.add_argument("--holdout", type=int, default=50_000)
    ap.add_argument("--features", type=int, default=1 << 20)
    ap.add_argument("--nnz", type=int, default=60)
    ap.add_argument("--candidates", type=int, default=40)
    ap.add_argument("--folds", type=int, default=5)
    ap.add_argument("--epochs", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=2048)
    ap.add_argument("--lr", type=float, default=0.2)
    ap.add_argument("--sklearn", action="store_true",
                    help="also fit sklearn LogisticRegression on the "
                         "same train split for a quality baseline")
    args = ap.parse_args()

    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    import torch

    sc = None
    if torch.cuda.is_available():
        from skdist_amd import Cluster

        sc = Cluster(require_gpu=True)

    t0 = time.perf_counter()
    X, y = make_text_csr(args.n + args.holdout, args.features, args.nnz,
                         seed=0)
    Xtr, ytr = X[: args.n], y[: args.n]
    Xte, yte = X[args.n:], y[args.n:]
    gen_s = time.perf_counter() - t0

    grid = {"C": list(np.logspace(-2, 2, args.candidates))}
    gs = DistGridSearchCV(
        LogisticRegression(epochs=args.epochs, momentum=0.0,
                           batch_size=args.batch_size, lr=args.lr,
                           random_state=0),
        grid, cv=args.folds, scoring="accuracy", sc=sc)
    t1 = time.perf_counter()
    gs.fit(Xtr, ytr)
    fit_s = time.perf_counter() - t1
    acc = float((gs.best_estimator_.predict(Xte) == yte).mean())
    n_fits = args.candidates * args.folds
    out = {
        "workload": "text-scale sparse DistGridSearchCV(LogisticRegression)",
        "n_docs": args.n, "n_features": args.features,
        "nnz_per_row": args.nnz, "candidates": args.candidates,
        "folds": args.folds, "epochs": args.epochs,
        "fit_wall_s": round(fit_s, 3),
        "fits_per_s": round(n_fits / fit_s, 2),
        "best_cv_score": round(float(gs.best_score_), 4),
        "holdout_acc": round(acc, 4),
        "gen_s": round(gen_s, 2),
        "batch_size": args.batch_size, "lr": args.lr,
        "device": "gpu" if sc is not None else "cpu",
    }
    if args.sklearn:
        from sklearn.linear_model import LogisticRegression as SkLR

        t2 = time.perf_counter()
        sk = SkLR(max_iter=100, C=float(gs.best_params_["C"])).fit(
            Xtr, ytr)
        out["sklearn_fit_s"] = round(time.perf_counter() - t2, 2)
        out["sklearn_holdout_acc"] = round(
            float((sk.predict(Xte) == yte).mean()), 4)
    line = json.dumps(out)
    print(line)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/r2_textscale.json", "a") as fh:
        fh.write(line + "\n")


if __name__ == "__main__":
    main()
