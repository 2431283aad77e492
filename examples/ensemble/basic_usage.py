"""
Distributed tree ensembles (reference analog:
examples/ensemble/basic_usage.py on breast_cancer; the reference reports
RF ROC-AUC 0.997 / ET 0.996).

With ``sc=Cluster()`` on a GPU node the trees grow through the batched
HIP histogram builder (LDS per-feature histograms, one level per kernel
sweep for a whole batch of trees); fitted trees come back as host numpy
arrays that pickle and predict with no GPU state.
"""

import numpy as np
from sklearn.datasets import load_breast_cancer
from sklearn.metrics import roc_auc_score

from skdist_amd.distribute.ensemble import (
    DistExtraTreesClassifier,
    DistRandomForestClassifier,
    DistRandomTreesEmbedding,
)


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()


X, y = load_breast_cancer(return_X_y=True)
X = np.asarray(X, dtype=np.float32)

for cls in (DistRandomForestClassifier, DistExtraTreesClassifier):
    clf = cls(n_estimators=100, random_state=0, sc=_sc())
    clf.fit(X, y)
    auc = roc_auc_score(y, clf.predict_proba(X)[:, 1])
    print(cls.__name__, "train ROC-AUC:", round(auc, 5))

emb = DistRandomTreesEmbedding(n_estimators=50, max_depth=5,
                               random_state=0, sc=_sc())
T = emb.fit_transform(X)
print("embedding shape:", T.shape)
