"""
Distribution primitives (reference: skdist/distribute/base.py:1-72).

``_clone`` mirrors sk-dist's sc-aware clone: estimator hyper-parameters are
deep-copied, but a scheduler handle (``sc``) held by a meta-estimator or its
nested estimator is pinned BY REFERENCE so cloning never duplicates or
pickles a live communicator (reference base.py:8-50).
"""

import copy

from sklearn.base import clone

from ..parallel.cluster import Cluster


def _is_cluster(obj):
    return isinstance(obj, Cluster)


def _clone(estimator, safe=True):
    """sklearn-style clone that carries any ``sc`` handle by reference."""
    sc = getattr(estimator, "sc", None)
    nested_sc = getattr(getattr(estimator, "estimator", None), "sc", None)
    if _is_cluster(sc):
        estimator.sc = None
    if _is_cluster(nested_sc):
        estimator.estimator.sc = None
    try:
        cloned = clone(estimator) if safe else copy.deepcopy(estimator)
    finally:
        if _is_cluster(sc):
            estimator.sc = sc
        if _is_cluster(nested_sc):
            estimator.estimator.sc = nested_sc
    if _is_cluster(sc):
        cloned.sc = sc
    if _is_cluster(nested_sc):
        cloned.estimator.sc = nested_sc
    return cloned


def _parse_partitions(partitions, n_tasks):
    """Resolve the ``partitions`` kwarg (reference base.py:53-64).

    In sk-dist this became Spark's ``numSlices``.  Our scheduler shards by
    task index, so partitions only caps the number of concurrent in-flight
    tasks per rank; 'auto' and None mean "one slot per task".
    """
    if partitions == "auto" or partitions is None:
        return n_tasks
    return int(partitions)


def _strip_sc(est):
    """Null scheduler handles before returning a fitted estimator
    (pickle contract, reference search.py:568-570, multiclass.py:283-285 —
    we keep the attribute as None so get_params/clone still work).
    """
    if getattr(est, "sc", None) is not None:
        est.sc = None
    nested = getattr(est, "estimator", None)
    if nested is not None and getattr(nested, "sc", None) is not None:
        nested.sc = None
    return est
