"""
Batched / distributed inference (reference: skdist/distribute/predict.py).

The reference wraps a fitted model into a Spark pandas UDF
(``get_prediction_udf`` predict.py:74-179) so executors score Arrow
batches.  The MI355X-native equivalents:

  * :func:`get_prediction_fn` — same marshalling contract (feature_type
    'numpy' | 'pandas' | 'text', column-wise inputs) but returns a plain
    vectorized callable usable anywhere (reference API parity);
  * :class:`DistPredictor` — streams an arbitrarily large row set through
    the cluster: rows are sharded contiguously across ranks, each rank
    scores its shard in chunks (GPU-resident GEMM for our native linear
    models, host predict otherwise), and results gather to every rank.
"""

import numpy as np
import pandas as pd

__all__ = ["get_prediction_fn", "DistPredictor"]


def _get_vals(cols, feature_type, names):
    """Column arrays -> model input (reference predict.py:59-71)."""
    if feature_type == "numpy":
        return np.column_stack([np.asarray(c) for c in cols])
    if feature_type == "pandas":
        if names is None:
            raise ValueError("names required for feature_type='pandas'")
        return pd.DataFrame(
            {n: np.asarray(c) for n, c in zip(names, cols)}
        )[list(names)]
    if feature_type == "text":
        return np.asarray(cols[0])
    raise ValueError(f"unknown feature_type: {feature_type!r}")


def get_prediction_fn(model, method="predict", feature_type="numpy",
                      names=None):
    """Vectorized prediction callable over column arrays
    (reference get_prediction_udf, predict.py:74-179).

    ``fn(*cols)`` returns a numpy array: predictions for 'predict', an
    [n, n_classes] probability matrix for 'predict_proba'.
    """
    if method not in ("predict", "predict_proba"):
        raise ValueError(f"unsupported method: {method!r}")

    def fn(*cols):
        vals = _get_vals(cols, feature_type, names)
        return getattr(model, method)(vals)

    return fn


class DistPredictor:
    """Stream batched inference over the GPU cluster.

    Args:
        model: any fitted sklearn-API estimator (pickled to every rank if
            they don't already hold it).
        sc: None (local) or a Cluster.
        method: 'predict' | 'predict_proba' | 'decision_function'.
        chunk_rows: rows per inference chunk.
    """

    def __init__(self, model, sc=None, method="predict", chunk_rows=1 << 20):
        self.model = model
        self.sc = sc
        self.method = method
        self.chunk_rows = chunk_rows

    def __call__(self, X):
        return self.predict(X)

    def predict(self, X):
        sc = self.sc
        if sc is None or not getattr(sc, "distributed", False):
            return self._predict_local(X)
        X = sc.sync_host_data(X)
        n = X.shape[0]
        # contiguous shard per rank
        per = (n + sc.world_size - 1) // sc.world_size
        lo = min(sc.rank * per, n)
        hi = min(lo + per, n)
        try:
            mine = self._predict_local(X[lo:hi]) if hi > lo else None
            err = None
        except Exception as e:  # noqa: BLE001 — re-raised post-gather on
            import traceback  # every rank (no stranded collective)

            mine = None
            err = f"{type(e).__name__}: {e}\n{traceback.format_exc()}"
        # gather ordered shards from all ranks
        import torch.distributed as dist

        from ..parallel.cluster import TaskFailedError

        boxes = [None] * sc.world_size
        dist.all_gather_object(boxes, (sc.rank, err, mine))
        for _, e, _b in boxes:
            if e is not None:
                raise TaskFailedError(
                    f"predict shard failed on a rank:\n{e}"
                )
        boxes = [b for _, _, b in sorted(boxes) if b is not None]
        return np.concatenate(boxes, axis=0)

    def _predict_local(self, X):
        fn = self._device_fn() or getattr(self.model, self.method)
        n = X.shape[0]
        if n <= self.chunk_rows:
            return np.asarray(fn(X))
        outs = []
        for lo in range(0, n, self.chunk_rows):
            outs.append(np.asarray(fn(X[lo : lo + self.chunk_rows])))
        return np.concatenate(outs, axis=0)

    def _device_fn(self):
        """GPU fast path: HIP-fitted forests score through the batched
        traversal kernel instead of the host loop (the device analog of
        the reference's executor-side predict, predict.py:160-178)."""
        if self.method not in ("predict", "predict_proba"):
            return None
        import torch

        if not torch.cuda.is_available():
            return None
        if getattr(self, "_flat", None) is None:
            from ..models.forest import flat_forest_for

            device = (
                self.sc.device
                if getattr(self.sc, "device", None) is not None
                else "cuda"
            )
            hook = getattr(self.model, "_device_predict_fn", None)
            if hook is not None:
                self._flat = hook(self.method, device) or False
            else:
                self._flat = flat_forest_for(self.model, device) or False
        if self._flat is False:
            return None
        if callable(self._flat) and not hasattr(self._flat, self.method):
            return self._flat
        return getattr(self._flat, self.method)
