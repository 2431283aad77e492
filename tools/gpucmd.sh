export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== full gpu suite =="
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1; echo "rc=$?"
tail -2 gpurun_out/pytest_gpu.log
echo "== bench (reverted kernels + fmask arg) =="
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_now.json | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s')"
echo "== rocprof: forest builder =="
cd /tmp
PYTHONPATH=/root/repo timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/proff -o proff -- python - > /root/repo/gpurun_out/proff.log 2>&1 <<'PY'
import numpy as np, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
n, f = 1_000_000, 64
X = rng.standard_normal((n, f)).astype(np.float32)
y = ((X @ rng.standard_normal(f)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, "cuda", is_cls=True)
b = ForestBuilder(ds, "gini", max_depth=10, max_features="sqrt", bootstrap=True, tree_batch=32)
trees = b.build(list(range(32)))
torch.cuda.synchronize()
PY
echo "rc=$?"
cd /root/repo
python tools/prof_summary.py gpurun_out/proff/proff_results.db 2>&1 | head -12 | tee profiles/r01_forest_1Mx64.txt
