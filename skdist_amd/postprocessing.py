"""
Postprocessing: voting over already-fitted estimators
(reference: skdist/postprocessing.py).

``SimpleVoter`` differs from sklearn's VotingClassifier in that it takes
FITTED estimators (fit happened elsewhere — typically distributed) and
only ensembles at predict time (reference postprocessing.py:17-42).
"""

import numpy as np
from sklearn.base import BaseEstimator, ClassifierMixin
from sklearn.preprocessing import LabelEncoder
from sklearn.utils import Bunch

__all__ = ["SimpleVoter"]


class SimpleVoter(ClassifierMixin, BaseEstimator):
    """Voting over pre-fitted (name, estimator) pairs.

    Args:
        estimators: list of fitted (name, estimator) tuples.
        classes: class labels (from any member estimator).
        voting: 'hard' (majority of predictions) or 'soft' (argmax of
            averaged probabilities).
        weights: optional per-estimator weights.
    """

    def __init__(self, estimators, classes, voting="hard", weights=None):
        self.estimators = estimators
        self.classes = classes
        self.voting = voting
        self.weights = weights
        self._assemble()

    def _assemble(self):
        names, clfs = zip(*self.estimators)
        self.estimators_ = clfs
        self.classes_ = np.asarray(self.classes)
        self.le_ = LabelEncoder()
        self.le_.classes_ = np.asarray(self.classes)

    @property
    def named_estimators(self):
        return Bunch(**dict(self.estimators))

    def _active_weights(self):
        if self.weights is None:
            return None
        return [
            w for (name, est), w in zip(self.estimators, self.weights)
            if est not in (None, "drop")
        ]

    def fit(self, X, y=None):
        """No-op fit (estimators are pre-fitted); refreshes attributes."""
        self._assemble()
        return self

    def predict(self, X):
        if self.voting == "soft":
            idx = self.predict_proba(X).argmax(axis=1)
        else:
            votes = np.column_stack(
                [self.le_.transform(clf.predict(X))
                 for clf in self.estimators_]
            )
            w = self._active_weights()
            k = len(self.classes_)
            counts = np.zeros((votes.shape[0], k))
            for j in range(votes.shape[1]):
                wj = 1.0 if w is None else w[j]
                counts[np.arange(votes.shape[0]), votes[:, j]] += wj
            idx = counts.argmax(axis=1)
        return self.le_.inverse_transform(idx)

    def predict_proba(self, X):
        if self.voting == "hard":
            raise AttributeError(
                f"predict_proba is not available when voting={self.voting!r}"
            )
        probas = np.asarray(
            [clf.predict_proba(X) for clf in self.estimators_]
        )
        return np.average(probas, axis=0, weights=self._active_weights())

    def transform(self, X):
        if self.voting == "soft":
            return np.asarray(
                [clf.predict_proba(X) for clf in self.estimators_]
            )
        return np.column_stack(
            [clf.predict(X) for clf in self.estimators_]
        )
