"""Same-box interleaved A/B of the dataset-prefetch overlap (round-1
methodology rule: box-to-box DVFS swamps small deltas)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression
from skdist_amd.models import linear as _lin

rng = np.random.default_rng(7)
X = rng.standard_normal((1_000_000, 256), dtype=np.float32)
w = rng.standard_normal(256).astype(np.float32) / 16
y = (X @ w > 0).astype(np.int64)
grid = {"C": list(np.logspace(-3, 3, 500))}
sc = Cluster(require_gpu=True)
saved = _lin._BatchedLinearBase.start_dataset_prefetch


def one():
    t0 = time.perf_counter()
    DistGridSearchCV(LogisticRegression(epochs=10, batch_size=8192,
                                        random_state=0),
                     grid, cv=5, scoring="accuracy", sc=sc).fit(X, y)
    import torch

    torch.cuda.synchronize()
    return time.perf_counter() - t0


one(); one()  # warm
a, b = [], []
for rep in range(6):
    _lin._BatchedLinearBase.start_dataset_prefetch = saved
    a.append(one())
    del _lin._BatchedLinearBase.start_dataset_prefetch
    b.append(one())
_lin._BatchedLinearBase.start_dataset_prefetch = saved
print("WITH prefetch   :", [round(t * 1000, 1) for t in a],
      "median", round(sorted(a)[len(a) // 2] * 1000, 1), "ms")
print("WITHOUT prefetch:", [round(t * 1000, 1) for t in b],
      "median", round(sorted(b)[len(b) // 2] * 1000, 1), "ms")
