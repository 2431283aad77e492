export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== tree_batch sweep (64 trees, 1M x 64, depth 12) =="
PYTHONPATH=/root/repo timeout 900 python - <<'PY' 2>&1 | tail -5
import numpy as np, time, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
y = ((X @ rng.standard_normal(f)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, "cuda", is_cls=True)
for tb in (16, 32, 64, 128):
    b = ForestBuilder(ds, "gini", max_depth=12, max_features="sqrt", bootstrap=True, tree_batch=tb)
    b.build([0])
    torch.cuda.synchronize(); t0 = time.time()
    trees = b.build(list(range(128)))
    torch.cuda.synchronize(); dt = time.time()-t0
    print(f"tree_batch={tb}: {dt:.2f}s = {128/dt:.1f} trees/s")
PY
echo "== rocprof forest (new balance) =="
cd /tmp
PYTHONPATH=/root/repo timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/proff2 -o proff2 -- python -c "
import numpy as np, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
X = rng.standard_normal((1_000_000, 64)).astype(np.float32)
y = ((X @ rng.standard_normal(64)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, 'cuda', is_cls=True)
ForestBuilder(ds, 'gini', max_depth=12, max_features='sqrt', bootstrap=True, tree_batch=32).build(list(range(32)))
torch.cuda.synchronize()" > /root/repo/gpurun_out/proff2.log 2>&1
cd /root/repo
python tools/prof_summary.py gpurun_out/proff2/proff2_results.db 2>&1 | head -9
echo "== full gpu suite + bench =="
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -1
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s')"
