export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /tmp
PYTHONPATH=/root/repo timeout 700 rocprofv3 --kernel-trace --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE -d /root/repo/gpurun_out/pmcf -o pmcf -- python -c "
import numpy as np, torch
from skdist_amd.models.forest import BinnedDataset, ForestBuilder
rng = np.random.default_rng(0)
X = rng.standard_normal((1_000_000, 64)).astype(np.float32)
y = ((X @ rng.standard_normal(64)) > 0).astype(np.int64)
ds = BinnedDataset(X, y, 'cuda', is_cls=True)
ForestBuilder(ds, 'gini', max_depth=12, max_features='sqrt', bootstrap=True, tree_batch=32).build(list(range(32)))
torch.cuda.synchronize()" > /root/repo/gpurun_out/pmcf.log 2>&1
echo "rc=$?"
