"""
DistMultiModelSearch (reference analog: the multi-model random search of
skdist/distribute/search.py:717-908): one task pool samples params
across heterogeneous model families and returns a per-model results
table.
"""

import numpy as np
from sklearn.datasets import load_breast_cancer

from skdist_amd.distribute.search import DistMultiModelSearch
from skdist_amd.models import LinearSVC, LogisticRegression


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = load_breast_cancer(return_X_y=True)
X = np.asarray(X, dtype=np.float32)

search = DistMultiModelSearch(
    models=[
        ("lr", LogisticRegression(epochs=20, random_state=0),
         {"C": [0.01, 0.1, 1.0, 10.0]}),
        ("svc", LinearSVC(epochs=20, random_state=0),
         {"C": [0.01, 0.1, 1.0]}),
    ],
    n=3, cv=3, sc=_sc(), random_state=0)
search.fit(X, y)
print(search.results_table_ if hasattr(search, "results_table_")
      else search.cv_results_)
print("best:", type(search.best_estimator_).__name__,
      round(search.best_score_, 4))
