#!/bin/bash
# Round-2 GPU call #6: measured text-voter pipeline + deep-tree inference
# probe (VERDICT weak #8: k_forest_predict unmeasured on deep trees).
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 900 python examples/postprocessing/text_voter.py 2>&1 | tee gpurun_out/r6_voter.log

timeout 600 python - <<'PYEOF' 2>&1 | tee gpurun_out/r6_deeptree.log
import time
import numpy as np
from skdist_amd import Cluster
from skdist_amd.distribute.ensemble import DistRandomForestClassifier
from skdist_amd.distribute.predict import DistPredictor

rng = np.random.default_rng(0)
X = rng.standard_normal((1_000_000, 64)).astype(np.float32)
w = rng.standard_normal(64)
y = ((X @ w + np.sin(X[:, 0] * 3) * 2 + 0.2 * rng.standard_normal(len(X))) > 0).astype(np.int64)
# deep, imbalanced trees: no depth cap, min_samples_leaf=1
m = DistRandomForestClassifier(n_estimators=64, max_depth=None,
                               random_state=0, sc=Cluster(require_gpu=True))
t0 = time.time(); m.fit(X[:500_000], y[:500_000]); fit_s = time.time() - t0
depths = []
for t in m.estimators_:
    # depth from tree arrays
    d = {0: 0}; mx = 0
    for i in range(len(t.feature)):
        if t.feature[i] >= 0:
            for ch in (t.left[i], t.right[i]):
                d[ch] = d[i] + 1
                mx = max(mx, d[ch])
    depths.append(mx)
print(f"fit {fit_s:.1f}s; depths min/med/max: {min(depths)}/{int(np.median(depths))}/{max(depths)}")
pred = DistPredictor(m, sc=Cluster(require_gpu=True), method="predict_proba")
t0 = time.time(); p = pred(X); dt = time.time() - t0
print(f"deep-tree predict_proba 1M x 64 trees: {dt:.2f}s = {len(X)/dt/1e6:.1f}M rows/s")
print("acc:", ((p[:, 1] > 0.5).astype(int) == y).mean())
PYEOF
