"""
DistFeatureEliminator (reference analog: examples/eliminate/basic_usage.py
— RF on 100k x 40 synthetic, the reference reports 182.7 s vs 8538.9 s
for serial RFECV, a 46.7x fan-out win).

Here every (feature-subset x fold) candidate is ONE COLUMN of a single
masked batched solve: removed features' weights are pinned to zero
per-column inside the SGD kernels, so the whole ladder scores in one
GPU pass instead of len(ladder) x folds separate fits.
"""

import numpy as np

from skdist_amd.distribute.eliminate import DistFeatureEliminator
from skdist_amd.models import LogisticRegression


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



rng = np.random.default_rng(0)
n, f = 20000, 40
X = rng.standard_normal((n, f)).astype(np.float32)
w = np.zeros(f)
w[:12] = rng.standard_normal(12) * 2
y = ((X @ w + 0.3 * rng.standard_normal(n)) > 0).astype(np.int64)
X[:, 12:] = rng.standard_normal((n, f - 12))

el = DistFeatureEliminator(
    LogisticRegression(epochs=15, random_state=0),
    sc=_sc(),  # Cluster() on a GPU node -> one masked batched solve
    min_features_to_select=8, step=4, cv=5)
el.fit(X, y)
print("kept features:", el.best_features_)
print("best CV score:", round(el.best_score_, 4))
