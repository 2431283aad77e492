"""Attribute per-fit wall time to engine phases (run on the GPU box).

Monkeypatches sync-bracketed timers around the flagship path's phases to
find where non-kernel time goes; informs bench optimization only.
"""

import os
import sys
import time
from collections import defaultdict

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import skdist_amd.models._sgd as sgd
import skdist_amd.models.linear as lin
from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression

ACC = defaultdict(float)
ON = {"flag": False}


def timed(name, fn):
    def wrap(*a, **k):
        if not ON["flag"]:
            return fn(*a, **k)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = fn(*a, **k)
        torch.cuda.synchronize()
        ACC[name] += time.perf_counter() - t0
        return out
    return wrap


sgd.DeviceDataset.__init__ = timed("dataset_build", sgd.DeviceDataset.__init__)
sgd.DeviceDataset.shuffled_views = timed(
    "shuffled_views", sgd.DeviceDataset.shuffled_views)
for mod in (sgd, lin):
    mod.batched_sgd_fit = timed("sgd_fit", mod.batched_sgd_fit)
    mod.batched_scores_by_fold = timed(
        "scores", mod.batched_scores_by_fold)


def main():
    rng = np.random.default_rng(7)
    n, f = 1_000_000, 256
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(f).astype(np.float32) / 16
    y = (X @ w > 0).astype(np.int64)
    grid = {"C": list(np.logspace(-3, 3, 500))}
    cluster = Cluster(require_gpu=True)

    def one():
        gs = DistGridSearchCV(
            LogisticRegression(epochs=10, batch_size=8192, random_state=0),
            grid, cv=5, scoring="accuracy", sc=cluster)
        gs.fit(X, y)
        return gs

    one()  # warmup
    ON["flag"] = True
    steps = 2
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one()
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    print(f"total/step: {total / steps * 1000:.1f} ms")
    covered = 0.0
    for k, v in sorted(ACC.items(), key=lambda kv: -kv[1]):
        print(f"  {k:16s} {v / steps * 1000:8.1f} ms/step")
        covered += v
    print(f"  {'(unattributed)':16s} {(total - covered) / steps * 1000:8.1f} ms/step")


if __name__ == "__main__":
    main()
