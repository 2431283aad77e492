"""
Large-fan-out grid search on digits (reference analog:
examples/search/hand_written_digits.py — the reference's HEADLINE
benchmark story: 750 fits (75 SVC candidates x cv=10) in 1.448 s wall
on a 640-core Spark cluster, ~302x aggregate-task-time fan-out;
BASELINE.md quotes it at ~517 fits/s).

Here the same 750-fit grid runs through our scheduler.  With the native
LogisticRegression and a GPU Cluster every (candidate x fold) model is
one COLUMN of a single batched MFMA solve — `bench.py` measures this at
9.4-11.8k fits/s on one MI355X at the 1Mx256 scale.  This living-doc
script keeps the reference's exact sklearn-SVC workload on the generic
task fan-out so it runs anywhere (sc=None -> local path).
"""

import time

import numpy as np
from sklearn.datasets import load_digits
from sklearn.svm import SVC

from skdist_amd.distribute.search import DistGridSearchCV

X, y = load_digits(return_X_y=True)

grid = {
    "C": [10.0 ** e for e in range(-4, 1)],          # 5
    "gamma": ["scale", "auto", 0.001, 0.01, 0.1],    # x5
    "kernel": ["rbf", "poly", "sigmoid"],            # x3 = 75 candidates
}
cv = 10  # 75 x 10 = 750 fits, the reference's headline task count

gs = DistGridSearchCV(SVC(), grid, cv=cv, sc=None, n_jobs=-1)
t0 = time.time()
gs.fit(X, y)
wall = time.time() - t0
n_fits = len(gs.cv_results_["params"]) * cv
task_time = float(
    np.sum(gs.cv_results_["mean_fit_time"]) * cv
    + np.sum(gs.cv_results_["mean_score_time"]) * cv
)
print(f"{n_fits} fits in {wall:.2f} s wall "
      f"({task_time:.1f} s aggregate task time)")
print("best:", gs.best_params_, "cv score:", round(gs.best_score_, 4))
