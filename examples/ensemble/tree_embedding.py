"""
DistRandomTreesEmbedding (reference analog:
examples/ensemble/tree_embedding.py — NB on circles data: 0.9734
transformed vs 0.4965 raw CV accuracy).
"""

import numpy as np
from sklearn.datasets import make_circles
from sklearn.model_selection import cross_val_score
from sklearn.naive_bayes import BernoulliNB

from skdist_amd.distribute.ensemble import DistRandomTreesEmbedding


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = make_circles(n_samples=5000, factor=0.5, noise=0.05,
                    random_state=0)
X = np.asarray(X, dtype=np.float32)

emb = DistRandomTreesEmbedding(n_estimators=50, max_depth=5,
                               random_state=0, sc=_sc())
T = emb.fit_transform(X)
nb_raw = cross_val_score(BernoulliNB(), X, y, cv=5).mean()
nb_emb = cross_val_score(BernoulliNB(), T, y, cv=5).mean()
print(f"NB raw CV acc {nb_raw:.4f} -> embedded {nb_emb:.4f} "
      f"(width {T.shape[1]})")
