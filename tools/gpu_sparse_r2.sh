#!/bin/bash
# Round-2 GPU call #2: sparse text-scale solver on hardware.
#   1. new GPU test files (sparse kernels vs eager mirror, boosting tier)
#   2. measured 1M x 2^20 hashed-text-shaped DistGridSearchCV (the
#      VERDICT round-2 target workload) -> gpurun_out/r2_textscale.json
#   3. rocprofv3 --stats of the sparse solve + the GBT fit
set -x
mkdir -p gpurun_out

timeout 600 python -m pytest tests/test_sparse_gpu.py tests/test_boosting_gpu.py -x -q \
    2>&1 | tee gpurun_out/r2_sparse_tests.log

timeout 900 python tools/sparse_scan_gpu.py \
    2>&1 | tee gpurun_out/r2_sparse_scan.log

timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 --sklearn \
    2>&1 | tee gpurun_out/r2_textscale.log

# rocprof: smaller run so the profile stays readable
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_sparse" -- \
    python "$GRAFT_REPO_ROOT/tools/textscale_bench.py" --n 200000 --holdout 20000 \
    --candidates 8 --folds 5 --epochs 10 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r2_sparse_prof.log" 2>&1

timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_boost" -- \
    python -c "
import numpy as np
from skdist_amd.models import HistGradientBoostingClassifier
rng = np.random.default_rng(0)
X = rng.standard_normal((500_000, 32)).astype(np.float32)
t = np.sin(X[:, 0]) + 0.5 * X[:, 1] ** 2 + X[:, 2]
y = (t > np.median(t)).astype(int)
import time; t0 = time.time()
m = HistGradientBoostingClassifier(n_estimators=100, random_state=0).fit(X, y)
print(f'GBT 100 trees on 500k x 32: {time.time()-t0:.2f}s')
print('train acc', (m.predict(X[:50000]) == y[:50000]).mean())
" > "$GRAFT_REPO_ROOT/gpurun_out/r2_boost_prof.log" 2>&1
