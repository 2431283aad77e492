#!/bin/bash
# Round-2 GPU call #9: new serving test, config-4 OvR probe, boosted
# search on GPU, post-optimization textscale profile, tier + bench.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r9_gpu_tier.log
timeout 600 python bench.py --gpus 1 --steps 5 --warmup 2 2>&1 | tee gpurun_out/r9_bench.log

# config-4 analog: 1000-class OvR on 5M x 512 (BASELINE config 4)
timeout 600 python - <<'PYEOF' 2>&1 | tee gpurun_out/r9_ovr1000.log
import time
import numpy as np
from skdist_amd import Cluster
from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
from skdist_amd.models import LinearSVC
rng = np.random.default_rng(0)
n, f, k = 5_000_000, 512, 1000
X = rng.standard_normal((n, f), dtype=np.float32)
W = rng.standard_normal((f, k)).astype(np.float32) / 16
y = (X[:100_000] @ W).argmax(axis=1)
y = np.concatenate([y, rng.integers(0, k, size=n - 100_000)])
t0 = time.time()
m = DistOneVsRestClassifier(LinearSVC(epochs=5, random_state=0),
                            sc=Cluster(require_gpu=True)).fit(X, y)
print(f"OvR 1000-class 5M x 512 fit: {time.time()-t0:.2f}s")
print("predict shape:", m.predict(X[:1000]).shape)
PYEOF

timeout 900 python examples/search/xgb.py 2>&1 | tee gpurun_out/r9_xgb.log

timeout 900 python - <<'PYEOF' 2>&1 | tee gpurun_out/r9_textscale_prof2.log
import cProfile, pstats, sys, os
sys.path.insert(0, os.getcwd())
sys.argv = ["x", "--n", "1000000", "--holdout", "50000", "--candidates",
            "40", "--folds", "5", "--epochs", "10", "--batch-size", "1024"]
from tools.textscale_bench import main
main()
pr = cProfile.Profile()
pr.enable(); main(); pr.disable()
pstats.Stats(pr).sort_stats("cumulative").print_stats(24)
PYEOF
