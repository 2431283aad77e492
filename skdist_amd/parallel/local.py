"""
Local (CPU) task fan-out — the ``sc=None`` escape hatch.

sk-dist runs every parallel axis through joblib when no sparkContext is given
(reference: skdist/distribute/search.py:388-409, multiclass.py:296-311,
ensemble.py:283-300, eliminate.py:168-186, encoder.py:130-134).  We keep the
same behavior so the plumbing works identically with and without GPUs.
"""

from joblib import Parallel, delayed


def run_local_tasks(task_fn, tasks, n_jobs=None, pre_dispatch="2*n_jobs"):
    """Run ``task_fn(task)`` for every task, returning results in task order.

    ``n_jobs=None`` runs sequentially in-process (cheapest for the small
    closures these meta-estimators produce); any other value delegates to
    joblib exactly like sklearn does.
    """
    if n_jobs is None or n_jobs == 1:
        return [task_fn(t) for t in tasks]
    return Parallel(n_jobs=n_jobs, pre_dispatch=pre_dispatch)(
        delayed(task_fn)(t) for t in tasks
    )
