"""
SimpleVoter (reference analog: examples/postprocessing/simple_voter.py —
vote over independently fitted models, each possibly trained
distributed; the reference reports a ~26x fan-out win on 20newsgroups).
"""

import numpy as np
from sklearn.datasets import load_breast_cancer
from sklearn.preprocessing import LabelEncoder

from skdist_amd.distribute.ensemble import DistRandomForestClassifier
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression
from skdist_amd.postprocessing import SimpleVoter


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = load_breast_cancer(return_X_y=True)
X = np.asarray(X, dtype=np.float32)

gs = DistGridSearchCV(LogisticRegression(epochs=30, random_state=0),
                      {"C": [0.1, 1.0, 10.0]}, cv=3, sc=_sc()).fit(X, y)
rf = DistRandomForestClassifier(n_estimators=100, random_state=0,
                                sc=_sc()).fit(X, y)
le = LabelEncoder().fit(y)
voter = SimpleVoter([("lr", gs), ("rf", rf)], classes=le.classes_,
                    voting="soft")
voter.fit(X, y)
print("voter train accuracy:", round((voter.predict(X) == y).mean(), 4))
