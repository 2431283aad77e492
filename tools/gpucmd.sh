export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== 10-step bench (allocator stability) =="
timeout 900 python bench.py --steps 10 --warmup 2 2>/dev/null | tail -1 | tee gpurun_out/bench_10step.json | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s')"
echo "== memory check across fits =="
PYTHONPATH=/root/repo timeout 300 python - <<'PY' 2>&1 | tail -3
import numpy as np, torch
from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import LogisticRegression
rng = np.random.default_rng(7)
X = rng.standard_normal((200_000, 64), dtype=np.float32)
y = (X[:, 0] > 0).astype(np.int64)
c = Cluster(require_gpu=True)
for i in range(6):
    DistGridSearchCV(LogisticRegression(epochs=5, random_state=0),
                     {"C": [0.1, 1.0]}, cv=3, sc=c).fit(X, y)
    if i in (0, 5):
        print(f"fit {i}: alloc {torch.cuda.memory_allocated()>>20} MiB, reserved {torch.cuda.memory_reserved()>>20} MiB")
PY
echo "== final steady profile (with k_score) =="
cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof6 -o prof6 -- python /root/repo/bench.py --steps 2 --warmup 1 > /root/repo/gpurun_out/prof6.log 2>&1
cd /root/repo
python tools/prof_summary.py gpurun_out/prof6/prof6_results.db 2>&1 | head -12
