"""cProfile the host side of one flagship fit (run on the GPU box)."""
import cProfile
import os
import pstats
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression

rng = np.random.default_rng(7)
X = rng.standard_normal((1_000_000, 256), dtype=np.float32)
y = (X @ (rng.standard_normal(256).astype(np.float32) / 16) > 0).astype(np.int64)
grid = {"C": list(np.logspace(-3, 3, 500))}
cluster = Cluster(require_gpu=True)


def one():
    gs = DistGridSearchCV(
        LogisticRegression(epochs=20, batch_size=8192, random_state=0),
        grid, cv=5, scoring="accuracy", sc=cluster)
    gs.fit(X, y)


one()  # warm
pr = cProfile.Profile()
pr.enable()
one()
one()
pr.disable()
st = pstats.Stats(pr)
st.sort_stats("cumulative").print_stats(28)
