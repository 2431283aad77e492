export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== forest gpu tests =="
timeout 900 python -m pytest tests/test_forest_gpu.py -x -q 2>&1 | tail -2
echo "== forest perf after host vectorization (was 24-26 trees/s) =="
PYTHONPATH=/root/repo timeout 900 python - <<'PY' 2>&1 | tail -4
import numpy as np, time, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
y = ((X @ rng.standard_normal(f)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, "cuda", is_cls=True)
for sub in (True, False):
    b = ForestBuilder(ds, "gini", max_depth=12, max_features="sqrt", bootstrap=True, tree_batch=32, subtract=sub)
    b.build([0])
    torch.cuda.synchronize(); t0 = time.time()
    trees = b.build(list(range(64)))
    torch.cuda.synchronize(); dt = time.time()-t0
    print(f"subtract={sub}: {dt:.2f}s = {64/dt:.2f} trees/s, nodes {np.mean([t.node_count for t in trees]):.0f}")
PY
echo "== config3 probe =="
PYTHONPATH=/root/repo timeout 600 python tools/config_probes.py forest 2>&1 | grep '^{'
