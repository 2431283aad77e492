#!/bin/bash
# Round-2 FIRST GPU call (SURVEY.md §8 queue item 6): validate the
# round-1 end-of-round CPU-only additions on hardware, then the full
# GPU tier.  Run via:
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash tools/gpu_validation_r2.sh'
set -x
mkdir -p gpurun_out

# 1. boosted family on the device path (reuses the forest kernels)
timeout 420 python - <<'EOF' 2>&1 | tee gpurun_out/r2_boost_gpu.log
import time
import numpy as np
from sklearn.model_selection import train_test_split

from skdist_amd.models import (
    HistGradientBoostingClassifier,
    HistGradientBoostingRegressor,
)

rng = np.random.default_rng(0)
X = rng.standard_normal((200_000, 32)).astype(np.float32)
t = (np.sin(X[:, 0]) + 0.5 * X[:, 1] ** 2 + X[:, 2]).astype(np.float64)
y = (t > np.median(t)).astype(int)
Xtr, Xte, ytr, yte = train_test_split(X, y, random_state=0)
t0 = time.time()
m = HistGradientBoostingClassifier(n_estimators=100, random_state=0).fit(Xtr, ytr)
fit_s = time.time() - t0
acc = (m.predict(Xte) == yte).mean()
print(f"GBT-cls GPU: fit {fit_s:.2f}s acc {acc:.4f}")
assert acc > 0.9, acc
r = HistGradientBoostingRegressor(n_estimators=60, random_state=0).fit(Xtr, t[: len(Xtr)])
print("GBT-reg GPU ok")
EOF

# 2. class_weight through the HIP row-weight plane
timeout 240 python - <<'EOF' 2>&1 | tee gpurun_out/r2_cw_gpu.log
import numpy as np
from sklearn.metrics import recall_score

from skdist_amd.models import LogisticRegression

rng = np.random.default_rng(0)
n = 500_000
X = rng.standard_normal((n, 64)).astype(np.float32)
y = (X[:, 0] * 2 - 2.8 + 0.5 * rng.standard_normal(n) > 0).astype(int)
m0 = LogisticRegression(epochs=10, random_state=0).fit(X, y)
m1 = LogisticRegression(epochs=10, class_weight="balanced", random_state=0).fit(X, y)
r0, r1 = recall_score(y, m0.predict(X)), recall_score(y, m1.predict(X))
print(f"recall plain {r0:.3f} balanced {r1:.3f}")
assert r1 > r0 + 0.05
EOF

# 3. sparse densify ingestion at GPU scale
timeout 240 python - <<'EOF' 2>&1 | tee gpurun_out/r2_sparse_gpu.log
import numpy as np
import scipy.sparse as sp

from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression
from skdist_amd import Cluster

rng = np.random.default_rng(0)
Xd = rng.standard_normal((200_000, 128)).astype(np.float32)
Xd[Xd < 0.8] = 0
X = sp.csr_matrix(Xd)
y = (Xd[:, 0] + Xd[:, 1] > 0.5).astype(np.int64)
gs = DistGridSearchCV(LogisticRegression(epochs=10, random_state=0),
                      {"C": [0.1, 1.0]}, cv=3, sc=Cluster(require_gpu=True))
gs.fit(X, y)
print("sparse GPU search best:", gs.best_score_)
assert gs.best_score_ > 0.8
EOF

# 4. full GPU tier
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tee gpurun_out/r2_gpu_tier.log
