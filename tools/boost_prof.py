"""GBT device-path profile workload (rocprofv3 target)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from skdist_amd.models import HistGradientBoostingClassifier

rng = np.random.default_rng(0)
X = rng.standard_normal((500_000, 32)).astype(np.float32)
t = np.sin(X[:, 0]) + 0.5 * X[:, 1] ** 2 + X[:, 2]
y = (t > np.median(t)).astype(int)
t0 = time.time()
m = HistGradientBoostingClassifier(n_estimators=100, random_state=0).fit(X, y)
print(f"GBT 100 trees on 500k x 32: {time.time() - t0:.2f}s")
print("train acc", (m.predict(X[:50_000]) == y[:50_000]).mean())
