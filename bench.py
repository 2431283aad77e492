"""
bench.py — flagship benchmark (driver contract).

Metric (BASELINE.json): candidate-fits/sec (whole node) for
DistGridSearchCV(LogisticRegression) with a 500-point C grid x 5-fold CV
on synthetic 1M x 256 tabular data (config 2), weak scaling: the grid grows
with the GPU count (500 candidates per GPU), so per-GPU work is fixed.

One "step" = one complete DistGridSearchCV.fit: device upload + broadcast,
batched MFMA-SGD training of all candidate x fold models, batched
test-fold scoring, cv_results_ assembly and the best-candidate refit.

The solver config (epochs=10, batch 8192) is the measured convergence
point at this data scale: best_score is identical (0.9199) for every
epoch count from 8 through 30 on the 1M x 256 task (sweep in
docs/BENCHMARKS.md) — 10 keeps a margin above the 8-epoch plateau.  The
per-step best_score is printed so reduced-quality runs are visible.

Run:  python bench.py [--gpus N] [--steps K] [--warmup W]
(N>1 is launched by the driver via torch.distributed.run, one rank/GPU.)
"""

import argparse
import json
import os
import time

import numpy as np


def make_synthetic(n, f, seed=0):
    """Random-init logistic ground truth; ~balanced binary labels."""
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f), dtype=np.float32)
    w = rng.standard_normal(f).astype(np.float32) / np.sqrt(f)
    logits = X @ w + 0.25 * rng.standard_normal(n).astype(np.float32)
    y = (logits > 0).astype(np.int64)
    return X, y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--candidates-per-gpu", type=int, default=500)
    ap.add_argument("--folds", type=int, default=5)
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--features", type=int, default=256)
    ap.add_argument("--epochs", type=int, default=10)
    ap.add_argument("--batch-size", type=int, default=8192)
    ap.add_argument("--dump-cv", default=None,
                    help="rank 0 saves the last step's mean_test_score "
                         "+ best_index_ to this .npz (dry-run equality "
                         "checks across world sizes)")
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    from skdist_amd import Cluster
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    cluster = Cluster(require_gpu=torch.cuda.is_available() is True)
    world = cluster.world_size
    rank = cluster.rank
    n_candidates = args.candidates_per_gpu * world
    n_fits = n_candidates * args.folds

    # every rank generates the identical dataset (no shared FS / network);
    # the engine still RCCL-broadcasts the device tensors from rank 0.
    X, y = make_synthetic(args.rows, args.features, seed=7)
    grid = {"C": list(np.logspace(-3, 3, n_candidates))}

    def one_step():
        est = LogisticRegression(
            epochs=args.epochs, batch_size=args.batch_size, random_state=0
        )
        gs = DistGridSearchCV(
            est, grid, cv=args.folds, scoring="accuracy", sc=cluster
        )
        gs.fit(X, y)
        return gs

    for _ in range(args.warmup):
        one_step()

    cluster.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    gs = None
    for _ in range(args.steps):
        gs = one_step()
    cluster.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if cluster.distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=cluster.device
                         if cluster.device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    value = n_fits * args.steps / elapsed
    if rank == 0 and args.dump_cv and gs is not None:
        np.savez(
            args.dump_cv,
            mean_test_score=gs.cv_results_["mean_test_score"],
            best_index=np.int64(gs.best_index_),
            coef=gs.best_estimator_.coef_,
        )
    if rank == 0:
        out = {
            "metric": "candidate-fits/sec",
            "value": value,
            "unit": "fits/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            # 517 fits/s = the reference's best published analog
            # (750 fits / 1.448 s on a 640-core Spark cluster,
            # BASELINE.md "hand_written_digits" row)
            "vs_baseline": value / 517.0,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "DistGridSearchCV(LogisticRegression-SGD)",
                "candidates": n_candidates,
                "folds": args.folds,
                "fits_per_step": n_fits,
                "n_samples": args.rows,
                "n_features": args.features,
                "epochs": args.epochs,
                "global_batch": args.batch_size,
                "seq_len": None,
                "parallelism": f"task-fanout-dp{world}",
                "best_score": None if gs is None else gs.best_score_,
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
