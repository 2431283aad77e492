export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== pytest gpu (forest) =="
timeout 900 python -m pytest tests/test_forest_gpu.py -x -q > gpurun_out/pytest_forest.log 2>&1; echo "rc=$?"
tail -12 gpurun_out/pytest_forest.log
echo "== pytest gpu (all) =="
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1; echo "rc=$?"
tail -3 gpurun_out/pytest_gpu.log
echo "== forest perf probe: 1M x 64, 64 trees =="
timeout 600 python - > gpurun_out/forest_perf.log 2>&1 <<'PY'
import numpy as np, time, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f)
y = ((X @ w + 0.3*rng.standard_normal(n)) > 0).astype(np.int64)
t0 = time.time()
ds = BinnedDataset(X, y, "cuda", is_cls=True)
torch.cuda.synchronize(); print("bin+upload", time.time()-t0)
b = ForestBuilder(ds, "gini", max_depth=12, max_features="sqrt", bootstrap=True, tree_batch=32)
t0 = time.time()
trees = b.build(list(range(32)))
torch.cuda.synchronize(); dt = time.time()-t0
print(f"32 trees depth12: {dt:.2f}s = {32/dt:.2f} trees/s, nodes {np.mean([t.node_count for t in trees]):.0f}")
t0 = time.time()
trees = b.build(list(range(64)))
torch.cuda.synchronize(); dt = time.time()-t0
print(f"64 trees depth12: {dt:.2f}s = {64/dt:.2f} trees/s")
from skdist_amd.models.forest import FlatForest
ff = FlatForest(trees, "cuda")
t0 = time.time(); p = ff.predict_value(X); torch.cuda.synchronize()
print(f"flat predict 1M rows x 64 trees: {time.time()-t0:.3f}s")
acc = (ds.classes_[p.argmax(1)] == y).mean(); print("train acc", acc)
PY
echo "perf rc=$?"
cat gpurun_out/forest_perf.log | tail -8
