"""
DistGridSearchCV basic usage (reference analog:
examples/search/basic_usage.py — LR C-grid on breast_cancer, cv=5,
roc_auc; the reference reports best CV 0.9925 at C=1.0).

On a GPU node, pass ``sc=Cluster()``: every (candidate x fold) model
trains as one column of a single batched MFMA-SGD solve against the
HBM-resident data.  With ``sc=None`` the same search runs the joblib-style
local path.  The fitted object strips all scheduler state and pickles
like a plain sklearn estimator.
"""

import pickle

import numpy as np
from sklearn.datasets import load_breast_cancer

from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression

data = load_breast_cancer()
X = np.asarray(data.data, dtype=np.float32)
y = data.target

import torch

sc = Cluster() if torch.cuda.is_available() else None
gs = DistGridSearchCV(
    LogisticRegression(epochs=30, random_state=0),
    {"C": [0.001, 0.01, 0.1, 1.0, 10.0, 100.0]},
    cv=5, scoring="roc_auc", sc=sc,
)
gs.fit(X, y)
print("best C:", gs.best_params_["C"], "best CV roc_auc:",
      round(gs.best_score_, 5))

blob = pickle.dumps(gs)           # sc-free, plain-sklearn pickle
model = pickle.loads(blob)
print("holdout proba shape:", model.predict_proba(X[:5]).shape)
