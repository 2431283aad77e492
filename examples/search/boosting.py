"""
Boosted-model search (reference analog: examples/search/xgb.py — the
reference rode an external xgboost through DistRandomizedSearchCV with
early stopping; this engine ships a native boosted family on its binned
histogram tree builder instead, so the same workflow needs no external
package).
"""

import numpy as np
from sklearn.datasets import load_breast_cancer

from skdist_amd.distribute.search import DistRandomizedSearchCV
from skdist_amd.models import HistGradientBoostingClassifier


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = load_breast_cancer(return_X_y=True)

search = DistRandomizedSearchCV(
    HistGradientBoostingClassifier(
        n_estimators=150, n_iter_no_change=8, random_state=0
    ),
    {
        "max_depth": [2, 3, 4],
        "learning_rate": [0.05, 0.1, 0.3],
        "subsample": [0.7, 1.0],
    },
    n_iter=4, cv=3, scoring="roc_auc", random_state=0, sc=_sc(),
)
search.fit(X.astype(np.float32), y)
print("best params:", search.best_params_)
print("best CV roc_auc:", round(search.best_score_, 5))
print("rounds used by the refit model:",
      search.best_estimator_.n_estimators_)
