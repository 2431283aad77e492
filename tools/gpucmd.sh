export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== forest gpu tests =="
timeout 600 python -m pytest tests/test_forest_gpu.py -x -q 2>&1 | tail -2
echo "== forest perf (same probe as r01: 32/64 trees depth12, 1M x 64) =="
PYTHONPATH=/root/repo timeout 600 python - <<'PY' 2>&1 | tail -4
import numpy as np, time, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder, FlatForest
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f)
y = ((X @ w + 0.3*rng.standard_normal(n)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, "cuda", is_cls=True)
b = ForestBuilder(ds, "gini", max_depth=12, max_features="sqrt", bootstrap=True, tree_batch=32)
b.build([0])  # warm
torch.cuda.synchronize(); t0 = time.time()
trees = b.build(list(range(64)))
torch.cuda.synchronize(); dt = time.time()-t0
print(f"64 trees depth12: {dt:.2f}s = {64/dt:.2f} trees/s (was 26.07)")
ff = FlatForest(trees, "cuda")
t0=time.time(); p = ff.predict_value(X); torch.cuda.synchronize()
print(f"predict 1M x 64 trees: {time.time()-t0:.3f}s")
print("acc", (ds.classes_[p.argmax(1)] == y).mean())
PY
