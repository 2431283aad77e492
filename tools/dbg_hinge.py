import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from skdist_amd import Cluster
from skdist_amd.distribute.multiclass import DistOneVsRestClassifier
from skdist_amd.models import LinearSVC

rng = np.random.default_rng(1)
n, f, k = 1_000_000, 128, 200
X = rng.standard_normal((n, f), dtype=np.float32)
W = rng.standard_normal((k, 32)).astype(np.float32)
y = (X[:, :32] @ W.T).argmax(axis=1)
for mom in (0.9, 0.0):
    for ep in (10, 30):
        ovr = DistOneVsRestClassifier(
            LinearSVC(epochs=ep, momentum=mom, random_state=0),
            sc=Cluster(require_gpu=True))
        ovr.fit(X, y)
        acc = float((ovr.predict(X[:50000]) == y[:50000]).mean())
        print(f"momentum={mom} epochs={ep}: acc {acc:.4f}")
