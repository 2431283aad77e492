#!/bin/bash
# Round-2 GPU call #14: scale-headroom probes + forest regression check.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

# config-3 probe: full 1024-tree forest on 10M x 64 (round-1: 35.2 s)
timeout 900 python - <<'PYEOF' 2>&1 | tee gpurun_out/r14_config3.log
import time
import numpy as np
from skdist_amd import Cluster
from skdist_amd.distribute.ensemble import DistRandomForestClassifier
rng = np.random.default_rng(0)
X = rng.standard_normal((10_000_000, 64), dtype=np.float32)
w = rng.standard_normal(64)
y = ((X @ w) > 0).astype(np.int64)
m = DistRandomForestClassifier(n_estimators=1024, random_state=0,
                               sc=Cluster(require_gpu=True))
t0 = time.time(); m.fit(X, y)
print(f"config-3: 1024 trees on 10M x 64 in {time.time()-t0:.1f}s "
      f"(round-1 evidence: 35.2 s)")
print("acc:", (m.predict(X[:100_000]) == y[:100_000]).mean())
PYEOF

timeout 900 python tools/textscale_bench.py --n 4000000 --holdout 100000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee gpurun_out/r14_text4m.log

timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --features 4194304 --candidates 40 --folds 5 --epochs 10 \
    --batch-size 1024 2>&1 | tee gpurun_out/r14_text4mfeat.log
