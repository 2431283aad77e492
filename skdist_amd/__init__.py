"""
skdist_amd — MI355X-native distributed meta-estimator engine.

A from-scratch rebuild of the capabilities of Ibotta/sk-dist
(reference: /root/reference/skdist/__init__.py:1-18) for a single node of
AMD Instinct MI355X GPUs.  Where sk-dist distributes independent estimator
fits (grid-search candidates, one-vs-rest binary problems, forest trees,
feature-elimination subsets, feature-union transformers) to a PySpark
cluster, this engine broadcasts the training data once over xGMI into the
HBM3E of every GPU (RCCL via torch.distributed) and batches/round-robins
the fit tasks across them, with hand-written CDNA4 HIP kernels doing the
actual solving (MFMA GEMMs for linear models, LDS histograms for trees).

The user-facing contract matches sk-dist:
  * every meta-estimator takes ``sc=`` — ``None`` for the local CPU path
    (sk-dist's joblib branch) or a :class:`skdist_amd.Cluster` for the GPU
    scheduler;
  * fitted estimators strip all scheduler/GPU state before returning, so
    they pickle and predict exactly like plain scikit-learn objects.
"""

__version__ = "0.1.0"

from .parallel.cluster import Cluster

__all__ = ["Cluster", "__version__"]
