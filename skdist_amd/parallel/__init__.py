"""
Execution backends: the local CPU path (sk-dist's ``sc=None`` joblib branch,
reference search.py:388-409) and the single-node GPU scheduler that replaces
Spark (reference SURVEY.md §2.3).
"""

from .cluster import Cluster
from .local import run_local_tasks

__all__ = ["Cluster", "run_local_tasks"]
