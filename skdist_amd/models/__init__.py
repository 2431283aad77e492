"""
GPU-native base estimators.

sk-dist delegates every FLOP to scikit-learn's native solvers (liblinear,
Cython trees — reference SURVEY.md §2.4).  These estimators replace them
with MI355X-first implementations: a *batched* mini-batch-SGD linear solver
(one MFMA-GEMM kernel sequence trains every candidate×fold×class model
simultaneously against HBM-resident data) and a histogram decision-tree
builder.  All of them keep the sklearn API and pickle with host numpy
weights only.
"""

from .boosting import (
    GradientBoostingClassifier,
    GradientBoostingRegressor,
    HistGradientBoostingClassifier,
    HistGradientBoostingRegressor,
)
from .linear import LinearSVC, LogisticRegression, Ridge

__all__ = [
    "LogisticRegression", "LinearSVC", "Ridge",
    "HistGradientBoostingClassifier", "HistGradientBoostingRegressor",
    "GradientBoostingClassifier", "GradientBoostingRegressor",
]
