"""Input-type parity with the sklearn estimator contract: pandas
DataFrames/Series, string class labels, integer regression targets, and
plain lists through the native models and meta-estimators (the reference
inherits all of this from sklearn; our solver normalizes labels on the
host before tensorizing — models/_sgd.py DeviceDataset).
"""

import numpy as np
import pandas as pd
import pytest

from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression, Ridge


@pytest.fixture(scope="module")
def frame_xy():
    rng = np.random.default_rng(0)
    Xd = rng.standard_normal((240, 6)).astype(np.float32)
    y_num = (Xd[:, 0] > 0).astype(int)
    y_str = np.where(y_num == 1, "pos", "neg")
    df = pd.DataFrame(Xd, columns=[f"f{i}" for i in range(6)])
    return Xd, df, y_num, y_str


def test_native_fit_pandas_string_labels(frame_xy):
    Xd, df, y_num, y_str = frame_xy
    m = LogisticRegression(epochs=10, random_state=0).fit(df, y_str)
    assert list(m.classes_) == ["neg", "pos"]
    assert (m.predict(df) == y_str).mean() > 0.95
    # identical geometry to the numeric-label fit
    m2 = LogisticRegression(epochs=10, random_state=0).fit(Xd, y_num)
    assert np.allclose(np.abs(m.coef_), np.abs(m2.coef_))


def test_search_pandas_series_labels(frame_xy):
    _, df, _, y_str = frame_xy
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.1, 1.0]}, cv=3,
    )
    gs.fit(df, pd.Series(y_str))
    assert gs.best_score_ > 0.9
    assert set(gs.predict(df.iloc[:5])) <= {"neg", "pos"}


def test_ovr_string_labels(frame_xy):
    Xd, _, _, _ = frame_xy
    y3 = np.array(["a", "b", "c"])[Xd[:, :3].argmax(axis=1)]
    ovr = DistOneVsRestClassifier(
        LogisticRegression(epochs=10, random_state=0)
    ).fit(Xd, y3)
    assert (ovr.predict(Xd) == y3).mean() > 0.9


def test_ridge_integer_targets_regress_on_values(frame_xy):
    """A regressor given integer y must fit the VALUES, never
    label-encoded indices (the task="reg" hint in DeviceDataset)."""
    Xd = frame_xy[0]
    y = (10 + 30 * (Xd[:, 0] > 0)).astype(np.int64)  # values 10 / 40
    r = Ridge(epochs=15, random_state=0).fit(Xd, y)
    pred = r.predict(Xd)
    assert abs(pred.mean() - y.mean()) < 3.0
    assert pred.std() > 5.0  # not collapsed to the 0..1 index range


def test_list_inputs(frame_xy):
    Xd, _, y_num, _ = frame_xy
    m = LogisticRegression(epochs=5, random_state=0).fit(
        Xd.tolist(), y_num.tolist()
    )
    assert (m.predict(Xd.tolist()) == y_num).mean() > 0.9
