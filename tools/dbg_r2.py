import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from skdist_amd.models._sgd import (ColumnSpec, DeviceDataset,
                                    batched_sgd_fit,
                                    batched_scores_by_fold)

rng = np.random.default_rng(0)
n, f = 12000, 16
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f)
yr = (X @ w + 0.2 * rng.standard_normal(n)).astype(np.float32)
ds = DeviceDataset(X, yr, device="cuda")
idx = np.arange(n)
splits = [(np.setdiff1d(idx, idx[k::3]), idx[k::3]) for k in range(3)]
ds.set_cv_partition(splits)
spec = ColumnSpec("cuda", col_fold=np.array([0, 1, 2], dtype=np.int32),
                  col_class=np.array([-1, -1, -1], dtype=np.int32),
                  col_lr=np.full(3, 0.5, np.float32),
                  col_l2=np.full(3, 1.25e-4, np.float32))
mf = np.array([0, 1, 2])
cc = np.array([-1, -1, -1], dtype=np.int32)
for mom in (0.0, 0.9):
    W = batched_sgd_fit(ds, spec, "squared", 10, 8192, seed=0,
                        momentum=mom)
    r = batched_scores_by_fold(ds, W, mf, cc, n_classes=2, metric="r2")
    print(f"momentum={mom}: r2 {np.round(r, 5)}")
