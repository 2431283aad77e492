"""
Batched inference (reference analog: examples/predict — the pandas-UDF
scoring path, predict.py:74-179).

``get_prediction_fn`` gives the same column-wise vectorized callable as
the reference's UDF factory; ``DistPredictor`` streams a huge frame
through the cluster — rows shard contiguously across ranks, and tree
ensembles (our HIP-fitted forests, sklearn forests, sklearn GBTs) score
through the device traversal kernel.
"""

import numpy as np
from sklearn.ensemble import GradientBoostingClassifier

from skdist_amd.distribute.predict import DistPredictor, get_prediction_fn


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



rng = np.random.default_rng(0)
X = rng.standard_normal((50000, 16)).astype(np.float32)
y = (X[:, 0] + X[:, 1] > 0).astype(int)
gbt = GradientBoostingClassifier(n_estimators=50, random_state=0)
gbt.fit(X[:5000], y[:5000])

fn = get_prediction_fn(gbt, method="predict_proba", feature_type="numpy")
print("udf-style output:", fn(*[X[:3, j] for j in range(16)]).shape)

pred = DistPredictor(gbt, sc=_sc(), method="predict_proba")
proba = pred(X)   # chunked; GPU traversal kernel when available
print("streamed proba:", proba.shape, "acc:",
      round((proba.argmax(1) == y).mean(), 4))
