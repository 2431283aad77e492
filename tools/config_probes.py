"""Measured probes for BASELINE.json configs 3-5 on ONE MI355X.

Each probe prints one JSON line; 8-GPU numbers follow from the task-
parallel structure (trees / classes / row-shards are independent).
Run on the GPU box: PYTHONPATH=/root/repo python tools/config_probes.py
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def sync():
    torch.cuda.synchronize()


def probe_forest(n=10_000_000, f=64, trees=128, depth=14):
    """Config 3 (scaled): DistRandomForestClassifier on 10M x 64."""
    from skdist_amd.models.forest import BinnedDataset, ForestBuilder

    rng = np.random.default_rng(0)
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(f).astype(np.float32)
    y = ((X @ w + 0.3 * rng.standard_normal(n).astype(np.float32)) > 0
         ).astype(np.int64)
    t0 = time.time()
    ds = BinnedDataset(X, y, "cuda", is_cls=True)
    sync()
    t_bin = time.time() - t0
    b = ForestBuilder(ds, "gini", max_depth=depth, max_features="sqrt",
                      bootstrap=True, tree_batch=32)
    t0 = time.time()
    built = b.build(list(range(trees)))
    sync()
    dt = time.time() - t0
    print(json.dumps({
        "probe": "config3_forest", "n": n, "f": f, "trees": trees,
        "max_depth": depth, "bin_s": round(t_bin, 2),
        "build_s": round(dt, 2), "trees_per_s": round(trees / dt, 2),
        "mean_nodes": int(np.mean([t.node_count for t in built])),
        "est_1024_trees_8gpu_s": round(1024 / (8 * trees / dt) + t_bin, 1),
    }))


def probe_ovr(n=5_000_000, f=512, k=1000, epochs=10):
    """Config 4 (1 GPU shard): OvR LinearSVC, 1000 binary columns."""
    from skdist_amd import Cluster
    from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
    from skdist_amd.models import LinearSVC

    rng = np.random.default_rng(1)
    X = rng.standard_normal((n, f), dtype=np.float32)
    W = rng.standard_normal((k, 32)).astype(np.float32)
    y = (X[:, :32] @ W.T).argmax(axis=1)
    ovr = DistOneVsRestClassifier(
        LinearSVC(epochs=epochs, random_state=0),
        sc=Cluster(require_gpu=True))
    t0 = time.time()
    ovr.fit(X, y)
    sync()
    dt = time.time() - t0
    acc = float((ovr.predict(X[:100000]) == y[:100000]).mean())
    print(json.dumps({
        "probe": "config4_ovr", "n": n, "f": f, "classes": k,
        "epochs": epochs, "fit_s": round(dt, 2),
        "binary_fits_per_s": round(k / dt, 2), "train_acc_100k": acc,
    }))


def probe_gbt_predict(rows=100_000_000, f=16):
    """Config 5 (scaled): bulk predict_proba of a fitted GBT through the
    device traversal kernel; extrapolate rows/s to the 1B-row frame."""
    from sklearn.ensemble import GradientBoostingClassifier

    from skdist_amd.models.forest import flat_forest_for

    rng = np.random.default_rng(2)
    Xtr = rng.standard_normal((20000, f)).astype(np.float32)
    ytr = (Xtr[:, 0] + Xtr[:, 1] > 0).astype(int)
    gbt = GradientBoostingClassifier(n_estimators=100, max_depth=3,
                                     random_state=0).fit(Xtr, ytr)
    flat = flat_forest_for(gbt, "cuda")
    X = rng.standard_normal((rows, f), dtype=np.float32)
    flat.predict_proba(X[:1_000_000])  # warm
    sync()
    t0 = time.time()
    out = flat.predict_proba(X)
    sync()
    dt = time.time() - t0
    print(json.dumps({
        "probe": "config5_gbt_predict", "rows": rows, "f": f,
        "trees": 100, "predict_s": round(dt, 2),
        "rows_per_s": int(rows / dt),
        "est_1e9_rows_8gpu_s": round(1e9 / (8 * rows / dt), 1),
        "check": float(out[:, 1].mean()),
    }))


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "forest"):
        probe_forest()
    if which in ("all", "ovr"):
        probe_ovr()
    if which in ("all", "gbt"):
        probe_gbt_predict()
