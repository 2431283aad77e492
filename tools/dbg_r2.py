import os, sys
sys.path.insert(0, "/root/repo")
import numpy as np
from sklearn.metrics import r2_score
from skdist_amd import Cluster
from skdist_amd.models import Ridge
import skdist_amd.models._sgd as sgd

rng = np.random.default_rng(0)
n, f = 12000, 16
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f)
yr = (X @ w + 0.2 * rng.standard_normal(n)).astype(np.float32)

# device-fitted single model, host-scored
m_dev = Ridge(epochs=10, random_state=0, sc=Cluster(require_gpu=True))
m_dev.fit(X[:8000], yr[:8000])
print("device-fit host-scored r2:", r2_score(yr[8000:], m_dev.predict(X[8000:])))

os.environ["SKDIST_AMD_ALLOW_EAGER"] = "1"
orig = sgd._use_hip
sgd._use_hip = lambda d: False
m_eag = Ridge(epochs=10, random_state=0, sc=Cluster(require_gpu=True))
m_eag.fit(X[:8000], yr[:8000])
print("eager-fit host-scored r2:", r2_score(yr[8000:], m_eag.predict(X[8000:])))
sgd._use_hip = orig

# scorer isolation: same device W, device-kernel r2 vs torch-path r2
from skdist_amd.models._sgd import ColumnSpec, DeviceDataset, batched_sgd_fit, batched_scores_by_fold
ds = DeviceDataset(X, yr, device="cuda")
splits = []
idx = np.arange(n)
for k in range(3):
    te = idx[k::3]; splits.append((np.setdiff1d(idx, te), te))
ds.set_cv_partition(splits)
spec = ColumnSpec("cuda", col_fold=np.array([0,1,2], dtype=np.int32),
                  col_class=np.array([-1,-1,-1], dtype=np.int32),
                  col_lr=np.full(3, 0.5, np.float32),
                  col_l2=np.full(3, 1e-4, np.float32))
W = batched_sgd_fit(ds, spec, "squared", 10, 8192, seed=0)
mf = np.array([0,1,2]); cc = np.array([-1,-1,-1], dtype=np.int32)
r_hip = batched_scores_by_fold(ds, W, mf, cc, n_classes=2, metric="r2")
sgd._use_hip = lambda d: False
r_tor = batched_scores_by_fold(ds, W, mf, cc, n_classes=2, metric="r2")
print("kernel r2:", r_hip)
print("torch  r2:", r_tor)
