import sys
sys.path.insert(0, "/root/repo")
import numpy as np
from skdist_amd import Cluster
from skdist_amd.distribute.search import DistGridSearchCV
from skdist_amd.models import Ridge

rng = np.random.default_rng(0)
n, f = 12000, 16
X = rng.standard_normal((n, f)).astype(np.float32)
w = rng.standard_normal(f)
yr = (X @ w + 0.2 * rng.standard_normal(n)).astype(np.float32)

for lr, ep in ((0.5, 10), (0.75, 10), (0.5, 20), (0.5, 40)):
    g = DistGridSearchCV(Ridge(epochs=ep, lr=lr, random_state=0),
                         {"alpha": [1.0]}, cv=3, scoring="r2",
                         sc=Cluster(require_gpu=True))
    g.fit(X, yr)
    print(f"device search lr={lr} epochs={ep}: r2 {g.best_score_:.5f}")
g = DistGridSearchCV(Ridge(epochs=10, random_state=0),
                     {"alpha": [1.0]}, cv=3, scoring="r2", sc=None)
g.fit(X, yr)
print("host-path search: r2", round(g.best_score_, 5))
