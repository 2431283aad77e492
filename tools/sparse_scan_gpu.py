"""GPU hyperparameter scan for the sparse text solver: single fits at
200k x 2^20 across (batch_size, lr, epochs) vs sklearn-lbfgs on the same
data.  Picks the sparse-path defaults with measured evidence."""

import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tools.textscale_bench import make_text_csr  # noqa: E402


def main():
    n, hold = 200_000, 20_000
    X, y = make_text_csr(n + hold, 1 << 20, 60, seed=0)
    Xtr, ytr = X[:n], y[:n]
    Xte, yte = X[n:], y[n:]

    from skdist_amd.models import LogisticRegression

    rows = []
    for bs, lr, ep in [
        (8192, 0.2, 10), (2048, 0.2, 10), (1024, 0.2, 10),
        (512, 0.2, 10), (512, 0.1, 10), (1024, 0.2, 30),
        (512, 0.2, 30), (2048, 0.5, 10), (256, 0.2, 10),
        (512, 0.4, 20),
    ]:
        t0 = time.perf_counter()
        m = LogisticRegression(
            epochs=ep, lr=lr, batch_size=bs, momentum=0.0,
            random_state=0).fit(Xtr, ytr)
        dt = time.perf_counter() - t0
        acc = (m.predict(Xte) == yte).mean()
        rows.append((bs, lr, ep, acc, dt))
        print(f"bs={bs:5d} lr={lr} ep={ep:2d}: acc {acc:.4f} "
              f"({dt:.2f}s)", flush=True)

    t0 = time.perf_counter()
    from sklearn.linear_model import LogisticRegression as SkLR

    sk = SkLR(max_iter=100).fit(Xtr, ytr)
    print(f"sklearn-lbfgs: acc {(sk.predict(Xte) == yte).mean():.4f} "
          f"({time.perf_counter() - t0:.1f}s)")


if __name__ == "__main__":
    main()
