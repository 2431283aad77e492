"""
Pipelines and searches compose both ways (reference analog:
examples/search/pipeline.py — pipeline-in-search 0.5975 vs
search-in-pipeline 0.5110 best CV F1w on 20newsgroups).
"""

import numpy as np
from sklearn.datasets import load_breast_cancer
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import StandardScaler

from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression


def _sc():
    """Cluster() on a GPU node, None for the local CPU path."""
    import torch

    if not torch.cuda.is_available():
        return None
    from skdist_amd import Cluster

    return Cluster()



X, y = load_breast_cancer(return_X_y=True)
X = np.asarray(X, dtype=np.float32)

# pipeline INSIDE the search: preprocessing is re-fit per fold (correct CV)
pipe_in_search = DistGridSearchCV(
    Pipeline([("scale", StandardScaler()),
              ("clf", LogisticRegression(epochs=20, random_state=0))]),
    {"clf__C": [0.1, 1.0, 10.0]}, cv=5, scoring="roc_auc", sc=_sc())
pipe_in_search.fit(X, y)
print("pipeline-in-search best CV:", round(pipe_in_search.best_score_, 5))

# search INSIDE a pipeline: the search sees pre-transformed features
search_in_pipe = Pipeline([
    ("scale", StandardScaler()),
    ("search", DistGridSearchCV(
        LogisticRegression(epochs=20, random_state=0),
        {"C": [0.1, 1.0, 10.0]}, cv=5, scoring="roc_auc", sc=_sc())),
])
search_in_pipe.fit(X, y)
print("search-in-pipeline best CV:",
      round(search_in_pipe.named_steps["search"].best_score_, 5))
