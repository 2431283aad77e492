#!/bin/bash
# Round-2 GPU call #12: production burn — repeated-fit VRAM stability,
# 1M-scale sparse kernel profile, graph-path soak.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 900 python - <<'PYEOF' 2>&1 | tee gpurun_out/r12_burn.log
import time
import numpy as np
import torch
from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression

rng = np.random.default_rng(7)
X = rng.standard_normal((1_000_000, 256), dtype=np.float32)
w = rng.standard_normal(256).astype(np.float32) / 16
y = (X @ w > 0).astype(np.int64)
grid = {"C": list(np.logspace(-3, 3, 500))}
mems, times = [], []
for i in range(12):
    t0 = time.perf_counter()
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, batch_size=8192, random_state=0),
        grid, cv=5, scoring="accuracy", sc=Cluster(require_gpu=True))
    gs.fit(X, y)
    torch.cuda.synchronize()
    times.append(time.perf_counter() - t0)
    mems.append(torch.cuda.memory_allocated() / 1e9)
    del gs
print("fit times:", [round(t, 3) for t in times])
print("allocated GB after each fit:", [round(m, 3) for m in mems])
assert max(mems) - min(mems) < 0.5, "VRAM growth across fits"
print(f"reserved GB: {torch.cuda.memory_reserved()/1e9:.2f}")
print("BURN OK")
PYEOF

PROF=/tmp/prof_out; mkdir -p "$PROF"
cd /tmp && export TMPDIR=/tmp
timeout 900 rocprofv3 --kernel-trace --stats -d "$PROF/sparse1m" -- \
    python "$GRAFT_REPO_ROOT/tools/textscale_bench.py" --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r12_sparse1m_prof.log" 2>&1
cd "$GRAFT_REPO_ROOT"
for db in "$PROF"/sparse1m/*/*.db; do
  python tools/prof_summary.py "$db" > gpurun_out/r12_sparse1m_summary.txt 2>&1
done
