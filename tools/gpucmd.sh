export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== 20-step bench =="
timeout 900 python bench.py --steps 20 --warmup 2 2>/dev/null | tail -1 | tee gpurun_out/bench_20step.json | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s')"
echo "== config3 FULL: 1024 trees on ONE GPU (10M x 64) =="
PYTHONPATH=/root/repo timeout 900 python - <<'PY' 2>&1 | tail -3
import numpy as np, time, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
n, f = 10_000_000, 64
X = rng.standard_normal((n, f), dtype=np.float32)
w = rng.standard_normal(f).astype(np.float32)
y = ((X @ w + 0.3*rng.standard_normal(n).astype(np.float32)) > 0).astype(np.int64)
t0=time.time(); ds = BinnedDataset(X, y, "cuda", is_cls=True); torch.cuda.synchronize()
tb = time.time()-t0
b = ForestBuilder(ds, "gini", max_depth=14, max_features="sqrt", bootstrap=True, tree_batch=32)
t0=time.time(); trees = b.build(list(range(1024))); torch.cuda.synchronize(); dt=time.time()-t0
print(f"bin {tb:.2f}s; 1024 trees on ONE GPU: {dt:.1f}s = {1024/dt:.1f} trees/s")
print(f"-> config 3 (1024 trees / 8 GPUs, 128 each): ~{dt/8 + tb:.1f}s measured-basis")
PY
echo "== config4 quality at 30 epochs =="
PYTHONPATH=/root/repo timeout 600 python tools/config_probes.py ovr 2>&1 | grep '^{' || true
