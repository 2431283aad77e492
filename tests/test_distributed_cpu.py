"""
Multi-process plumbing tests: the SPMD scheduler over gloo with
world_size=2 on CPU — the stand-in for the 8-GPU RCCL path (SURVEY.md §4:
"host-simulated scheduler mode standing in for pytest-spark's local mode").
"""

import multiprocessing as mp
import os
import pickle
import socket

import numpy as np
import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _spmd_worker(rank, world_size, port, path, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        result = _SCENARIOS[path](rank)
        q.put((rank, "ok", result))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "err", f"{e}\n{traceback.format_exc()}"))
    finally:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()


def _scenario_batched(rank):
    from sklearn.datasets import load_breast_cancer

    from skdist_amd import Cluster
    from skdist_amd.distribute.search import DistGridSearchCV
    from skdist_amd.models import LogisticRegression

    sc = Cluster()
    # rank 0 holds the data; other ranks receive it via broadcast
    X = y = None
    if rank == 0:
        X, y = load_breast_cancer(return_X_y=True)
    gs = DistGridSearchCV(
        LogisticRegression(epochs=10, random_state=0),
        {"C": [0.01, 0.1, 1.0, 10.0]},
        cv=3, scoring="roc_auc", sc=sc,
    )
    gs.fit(X, y)
    blob = pickle.dumps(gs)  # sc stripped, must pickle
    return {
        "best_score": gs.best_score_,
        "scores": list(gs.cv_results_["mean_test_score"]),
        "pickle_len": len(blob),
    }


def _scenario_generic(rank):
    from sklearn.datasets import load_iris
    from sklearn.linear_model import LogisticRegression as SkLogReg

    from skdist_amd import Cluster
    from skdist_amd.distribute.search import DistGridSearchCV

    sc = Cluster()
    X = y = None
    if rank == 0:
        X, y = load_iris(return_X_y=True)
    gs = DistGridSearchCV(
        SkLogReg(solver="liblinear"), {"C": [0.1, 1.0, 10.0]}, cv=3, sc=sc
    )
    gs.fit(X, y)
    return {
        "best_score": gs.best_score_,
        "scores": list(gs.cv_results_["mean_test_score"]),
    }


_SCENARIOS = {
    "batched": _scenario_batched,
    "generic": _scenario_generic,
}


def _run_spmd(scenario, world_size=2):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(
            target=_spmd_worker, args=(r, world_size, port, scenario, q)
        )
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(world_size):
        rank, status, payload = q.get(timeout=240)
        assert status == "ok", f"rank {rank}: {payload}"
        outs[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return outs


@pytest.mark.timeout(300)
def test_spmd_batched_gloo():
    outs = _run_spmd("batched")
    assert outs[0]["best_score"] > 0.98
    # every rank assembled identical results
    assert np.allclose(outs[0]["scores"], outs[1]["scores"])


@pytest.mark.timeout(300)
def test_spmd_generic_gloo():
    outs = _run_spmd("generic")
    assert outs[0]["best_score"] > 0.9
    assert np.allclose(outs[0]["scores"], outs[1]["scores"])
