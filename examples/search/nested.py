"""
Nested meta-estimators (reference analog: examples/search/nested.py —
DistGridSearchCV wrapped around DistOneVsRestClassifier, and
OvR-of-search; the reference compares nested CV scores).

Nesting works in either direction because every Dist* estimator follows
the sklearn estimator protocol and strips its scheduler before pickling.
"""

import numpy as np
from sklearn.datasets import make_classification

from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = make_classification(
    n_samples=2000, n_features=20, n_informative=10, n_classes=4,
    random_state=0)
X = np.asarray(X, dtype=np.float32)

# search over the inner binary estimator's C, OvR outside
ovr_of_search = DistOneVsRestClassifier(
    DistGridSearchCV(
        LogisticRegression(epochs=15, random_state=0),
        {"C": [0.1, 1.0, 10.0]}, cv=3, sc=_sc()),
    sc=_sc())
ovr_of_search.fit(X, y)
print("OvR(search) acc:", round((ovr_of_search.predict(X) == y).mean(), 4))

# search over OvR as a whole (C reaches the inner estimator via set_params)
search_of_ovr = DistGridSearchCV(
    DistOneVsRestClassifier(LogisticRegression(epochs=15, random_state=0)),
    {"estimator__C": [0.1, 1.0, 10.0]}, cv=3, sc=_sc())
search_of_ovr.fit(X, y)
print("search(OvR) best CV:", round(search_of_ovr.best_score_, 4))
