"""Boosted-model search (reference analog: examples/search/xgb.py —
XGBClassifier/XGBRegressor 54-candidate grids, cv=5: best clf roc_auc
0.99369, best reg score -18.452).

The reference treated xgboost as a pass-through: any sklearn-API
estimator rides the task fan-out.  That contract holds here — if
xgboost is installed it is used verbatim; otherwise the engine's
native hist-GBT family (models/boosting.py, GradientBoosting* API on
the HIP binned tree builder) runs the same grids."""

import numpy as np
from sklearn.datasets import load_breast_cancer, load_diabetes

from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV

try:
    from xgboost import XGBClassifier, XGBRegressor

    clf = XGBClassifier(eval_metric="logloss", use_label_encoder=False)
    reg = XGBRegressor()
    print("using xgboost (pass-through estimator on the fan-out)")
except ImportError:
    from skdist_amd.models import (
        GradientBoostingClassifier,
        GradientBoostingRegressor,
    )

    clf = GradientBoostingClassifier(random_state=0)
    reg = GradientBoostingRegressor(random_state=0)
    print("xgboost not installed: using the native hist-GBT family")

import torch

sc = Cluster() if torch.cuda.is_available() else None


if torch.cuda.is_available():
    grid = {
        "learning_rate": [0.05, 0.1, 0.2],
        "max_depth": [3, 4, 5],
        "n_estimators": [50, 100],
        "subsample": [0.8, 1.0],
    }  # 36 candidates x cv=5 = 180 boosted fits on the fan-out
else:  # small CPU demo grid
    grid = {
        "learning_rate": [0.1, 0.2],
        "max_depth": [3],
        "n_estimators": [30],
    }

data = load_breast_cancer()
gs = DistGridSearchCV(clf, grid, cv=5, scoring="roc_auc", sc=sc)
gs.fit(np.asarray(data.data, dtype=np.float32), data.target)
print("clf best roc_auc:", round(gs.best_score_, 5), gs.best_params_)

dia = load_diabetes()
sc2 = Cluster() if torch.cuda.is_available() else None
gr = DistGridSearchCV(reg, grid, cv=5,
                      scoring="neg_mean_squared_error", sc=sc2)
gr.fit(np.asarray(dia.data, dtype=np.float32), dia.target)
print("reg best neg-MSE:", round(gr.best_score_, 3), gr.best_params_)
