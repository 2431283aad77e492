"""
GPU-native linear estimators with the sklearn API.

These replace the liblinear/lbfgs solvers sk-dist leaned on (reference
SURVEY.md §2.4 row 1).  ``fit`` trains one model; ``batched_cv_fit_score``
is the protocol hook :class:`skdist_amd.distribute.search.DistBaseSearchCV`
uses to train EVERY (candidate × fold) model in one batched solve on the
GPU cluster.

Fitted state is host numpy only (``coef_``, ``intercept_``, ``classes_``)
— models pickle and predict exactly like sklearn estimators (sk-dist
contract, reference search.py:568-570).
"""

import time
from itertools import combinations

import numpy as np
from sklearn.base import BaseEstimator, ClassifierMixin, RegressorMixin

from ._sgd import (
    LOSS_HINGE,
    LOSS_LOG,
    LOSS_SQUARED,
    ColumnSpec,
    DeviceDataset,
    batched_scores_by_fold,
    batched_sgd_fit,
)


class FallbackToGeneric(Exception):
    """Raised when a batched solve cannot honor the request; the search
    falls back to the per-task generic path."""


def _make_dataset(X, y, cluster=None, device=None, standardize=True,
                  sample_weight=None, task=None, classes=None):
    """DeviceDataset (dense MFMA path) or SparseDeviceDataset (CSR
    text-scale path).  Sparse X stays sparse when densifying would be
    unreasonable (> SKDIST_AMD_SPARSE_GB dense, default 2 GB, or wider
    than 2^16 features — hashed-text shapes); smaller sparse inputs
    densify onto the faster MFMA path as before.  SKDIST_AMD_FORCE_SPARSE=1
    forces the sparse path (tests)."""
    import os as _os

    import scipy.sparse as _sp

    if _sp.issparse(X):
        force = _os.environ.get("SKDIST_AMD_FORCE_SPARSE") == "1"
        gb = X.shape[0] * X.shape[1] * 4.0 / 1e9
        wide = X.shape[1] > 65536
        limit = float(_os.environ.get("SKDIST_AMD_SPARSE_GB", "2"))
        if force or wide or gb > limit:
            from ._sparse_sgd import SparseDeviceDataset

            return SparseDeviceDataset(
                X, y, cluster=cluster, device=device,
                sample_weight=sample_weight, task=task, classes=classes)
    return DeviceDataset(
        X, y, cluster=cluster, device=device, standardize=standardize,
        sample_weight=sample_weight, task=task, classes=classes)


_DEVICE_METRICS = {
    "accuracy", "f1", "f1_weighted", "f1_macro", "neg_log_loss", "roc_auc",
    "r2", "neg_mean_squared_error",
}


class _DatasetPrefetch:
    """Background DeviceDataset build so the H2D upload + standardize
    kernels overlap the host-side cv.split work in DistBaseSearchCV.fit
    (~30 ms each at flagship scale — independent by construction).

    Collective safety: the build may issue collectives (the SPMD
    have-flags exchange / broadcast); every rank starts the thread at
    the same program point and the caller MUST ``wait()`` before issuing
    any other collective (DistBaseSearchCV does, on every path)."""

    def __init__(self, build_fn):
        import threading

        self._result = None
        self._exc = None

        def run():
            try:
                self._result = build_fn()
            except BaseException as e:  # re-raised at get()
                self._exc = e

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()

    def wait(self):
        self._thread.join()

    def get(self):
        self.wait()
        if self._exc is not None:
            raise self._exc
        return self._result


class _BatchedLinearBase(BaseEstimator):
    """Shared machinery for the SGD-trained linear family."""

    _loss = LOSS_LOG  # overridden

    def start_dataset_prefetch(self, X, y, cluster):
        """Kick off the device dataset build on a worker thread (search
        hook; sample_weight searches never reach the batched path so the
        prefetch only covers the plain case)."""
        is_clf = isinstance(self, ClassifierMixin)

        def build():
            sw = self._merged_sample_weight(y, None)
            return _make_dataset(
                X, y, cluster=cluster, standardize=self.standardize,
                sample_weight=sw, task="cls" if is_clf else "reg")

        return _DatasetPrefetch(build)

    def _lam(self, n_train):
        """Per-column L2 coefficient from the sklearn-style hyper-param."""
        C = getattr(self, "C", None)
        if C is not None:
            return 1.0 / (float(C) * max(n_train, 1))
        return float(getattr(self, "alpha", 1.0)) / max(n_train, 1)

    def _hyper_names(self):
        return {"C"} if hasattr(self, "C") else {"alpha"}

    # ------------------------------------------------------------------ #
    def _merged_sample_weight(self, y, sample_weight):
        """Fold ``class_weight`` ('balanced' | dict) into per-row weights
        (the solver's fused row-weight plane carries both)."""
        cw = getattr(self, "class_weight", None)
        if cw is None or y is None:
            return sample_weight
        from sklearn.utils.class_weight import compute_sample_weight

        w = compute_sample_weight(cw, np.asarray(y))
        if sample_weight is not None:
            w = w * np.asarray(sample_weight, dtype=np.float64)
        return w.astype(np.float32)

    def fit(self, X, y, sample_weight=None):
        t0 = time.perf_counter()
        sc = getattr(self, "sc", None)
        is_clf = isinstance(self, ClassifierMixin)
        sample_weight = self._merged_sample_weight(y, sample_weight)
        ds = _make_dataset(
            X, y,
            cluster=None,  # single fit: local device, no broadcast
            device=sc.device if sc is not None else None,
            standardize=self.standardize,
            sample_weight=sample_weight,
            task="cls" if is_clf else "reg",
        )
        ds.set_cv_partition([])  # no CV mask
        if is_clf:
            k = len(ds.classes_)
            if k < 2:
                raise ValueError(
                    "This solver needs samples of at least 2 classes in "
                    f"the data, but the data contains only one class: "
                    f"{ds.classes_[0]!r}"
                )
            ncols = 1 if k == 2 else k
            col_class = (
                np.array([1], dtype=np.int32) if k == 2
                else np.arange(k, dtype=np.int32)
            )
        else:
            ncols = 1
            col_class = np.array([-1], dtype=np.int32)
        lam = self._lam(ds.n)
        spec = ColumnSpec(
            ds.device,
            col_fold=np.full(ncols, -2, dtype=np.int32),
            col_class=col_class,
            col_lr=np.full(ncols, self.lr, dtype=np.float32),
            col_l2=np.full(ncols, lam, dtype=np.float32),
        )
        W = batched_sgd_fit(
            ds, spec, self._loss, self.epochs, self.batch_size,
            seed=self._seed(), momentum=self.momentum,
            lr_decay=getattr(self, "lr_decay", 0.0),
            adaptive=getattr(self, "adaptive", None),
        )
        self._store_fitted(ds, W, is_clf)
        self.n_features_in_ = ds.f
        self.fit_time_ = time.perf_counter() - t0
        return self

    def _seed(self):
        rs = getattr(self, "random_state", None)
        return 0 if rs is None else int(rs)

    def _store_fitted(self, ds, W, is_clf):
        Wh = W if isinstance(W, np.ndarray) else W.cpu().numpy()
        # [fa(+pad), ncols]
        w_std = Wh[: ds.f].T       # [ncols, f] standardized space
        b_std = Wh[ds.intercept_row]
        w_raw = np.empty_like(w_std)
        b_raw = np.empty_like(b_std)
        for c in range(w_std.shape[0]):
            w_raw[c], b_raw[c] = ds.unstandardize_coef(w_std[c], b_std[c])
        if is_clf:
            self.classes_ = ds.classes_
            self.coef_ = w_raw
            self.intercept_ = b_raw
        else:
            self.coef_ = w_raw[0]
            self.intercept_ = float(b_raw[0])

    # ------------------------------------------------------------------ #
    # host-side inference (numpy, sklearn-compatible)
    # ------------------------------------------------------------------ #
    def decision_function(self, X):
        import scipy.sparse as _sp

        if not hasattr(self, "coef_"):
            from sklearn.exceptions import NotFittedError

            raise NotFittedError(
                f"This {type(self).__name__} instance is not fitted yet. "
                "Call 'fit' before using this estimator.")

        if _sp.issparse(X):
            scores = (
                np.asarray(X.astype(np.float64) @ self.coef_.T)
                + self.intercept_
            )
        else:
            X = np.asarray(X, dtype=np.float64)
            scores = X @ self.coef_.T + self.intercept_
        if scores.ndim == 2 and scores.shape[1] == 1:
            return scores.ravel()
        return scores

    # ------------------------------------------------------------------ #
    # the batched-search protocol (used by DistBaseSearchCV.fit)
    # ------------------------------------------------------------------ #
    def batched_cv_fit_score(self, X, y, candidate_params, cv_splits,
                             scoring, scorers, cluster,
                             return_train_score=False,
                             sample_weight=None, prefetch=None):
        """Train all (candidate × fold) models in one batched device solve.

        Returns the same per-task result dicts the generic path produces
        (task order: candidates outer, folds inner) so ``_format_results``
        is shared.  Raises :class:`FallbackToGeneric` when the request
        can't be batched (unsupported params / scoring / non-partition CV).
        """
        if return_train_score:
            raise FallbackToGeneric("return_train_score not batched yet")
        if sample_weight is not None:
            # sklearn forwards an explicit fit-param sample_weight to the
            # SCORERS too (test-sliced); the device scorers are
            # unweighted, so the generic path keeps exact parity.
            # (class_weight stays batched — sklearn never weights
            # scoring for it.)
            raise FallbackToGeneric(
                "sample_weight weights the test scores in sklearn: "
                "generic path")
        metric = self._device_metric(scoring)
        # single comprehension: the set().union(*(set(p) ...)) form
        # measured 40 ms/fit at 500 candidates (profiles/r4 host trace)
        unsupported = {
            k for p in candidate_params for k in p
        } - self._hyper_names() - {"lr"}
        if unsupported:
            raise FallbackToGeneric(f"non-batchable params {unsupported}")

        is_clf = isinstance(self, ClassifierMixin)
        t0 = time.perf_counter()
        # class_weight folds into the row-weight plane.  sklearn computes
        # 'balanced' from each TRAINING fold's y; the plane is one weight
        # per row shared by every column, so the batched path uses the
        # full-y weights — exact for stratified folds, and guarded here:
        # if any fold's class ratios deviate >1% from the full data the
        # solve falls back to the generic per-task path (which matches
        # sklearn exactly) instead of silently diverging.
        self._check_balanced_foldable(y, cv_splits)
        if prefetch is not None:
            # built concurrently with the caller's cv.split work
            # (sample_weight searches never reach here: guarded above)
            ds = prefetch.get()
        else:
            sw = self._merged_sample_weight(y, sample_weight)
            ds = _make_dataset(
                X, y,
                cluster=cluster, standardize=self.standardize,
                sample_weight=sw,
                task="cls" if is_clf else "reg",
            )
        if not ds.set_cv_partition(cv_splits):
            raise FallbackToGeneric("cv splits do not partition the data")
        n_folds = len(cv_splits)
        n_cand = len(candidate_params)
        fold_train_n = [len(tr) for tr, _ in cv_splits]
        fold_test_n = [len(te) for _, te in cv_splits]

        if is_clf and ds.classes_ is not None and len(ds.classes_) >= 2:
            # a training fold missing one of the global classes makes
            # sklearn raise in that task (error_score semantics); the
            # joint batched solve would instead silently train a
            # degenerate column — defer to the exact generic path.
            import torch as _t

            k_all = len(ds.classes_)
            flat = ds.fold_id.to(_t.int64) * k_all + ds.y_int.to(_t.int64)
            fold_counts = _t.bincount(
                flat, minlength=n_folds * k_all
            ).reshape(n_folds, k_all)
            train_counts = fold_counts.sum(dim=0, keepdim=True) - fold_counts
            if bool((train_counts == 0).any()):
                raise FallbackToGeneric(
                    "a training fold is missing a class: generic path "
                    "(sklearn error_score semantics)")

        if is_clf:
            k = len(ds.classes_)
            n_classes = k
            cols_per_model = 1 if k == 2 else k
            cls = (
                np.array([1], dtype=np.int32) if k == 2
                else np.arange(k, dtype=np.int32)
            )
            if metric == "roc_auc" and k > 2:
                raise FallbackToGeneric(
                    "multiclass roc_auc needs sklearn's multi_class "
                    "semantics: generic path")
        else:
            n_classes = 2
            cols_per_model = 1
            cls = np.array([-1], dtype=np.int32)

        # this rank's shard of candidates
        cand_ids = (
            cluster.shard_indices(n_cand) if cluster is not None
            else list(range(n_cand))
        )
        col_fold, col_class, col_lr, col_l2, col_model = [], [], [], [], []
        model_folds = []   # per model: its test fold, -2 for full-data
        local_tasks = []   # (task_id, model_idx, fold)
        full_models = {}   # cand_id -> model_idx of the full-data model
        model_idx = 0
        self_params = self.get_params()
        for ci in cand_ids:
            params = candidate_params[ci]
            lr = float(params.get("lr", self.lr))
            merged = {**self_params, **params}
            for fold in range(n_folds):
                lam = _lam_from(merged, fold_train_n[fold])
                for cc in cls:
                    col_fold.append(fold)
                    col_class.append(cc)
                    col_lr.append(lr)
                    col_l2.append(lam)
                    col_model.append(model_idx)
                model_folds.append(fold)
                local_tasks.append((ci * n_folds + fold, model_idx, fold))
                model_idx += 1
            # one extra full-data model per candidate: trains alongside the
            # fold models so the search's "refit" is a column extraction
            # instead of a second solve (MI355X-batched replacement for
            # reference search.py:543-550's driver-side refit)
            lam = _lam_from(merged, ds.n)
            for cc in cls:
                col_fold.append(-2)  # trains on every row
                col_class.append(cc)
                col_lr.append(lr)
                col_l2.append(lam)
                col_model.append(model_idx)
            model_folds.append(-2)
            full_models[ci] = model_idx
            model_idx += 1

        results = {}
        refit_local = {}
        if model_idx > 0:
            spec = ColumnSpec(
                ds.device,
                col_fold=np.asarray(col_fold, dtype=np.int32),
                col_class=np.asarray(col_class, dtype=np.int32),
                col_lr=np.asarray(col_lr, dtype=np.float32),
                col_l2=np.asarray(col_l2, dtype=np.float32),
            )
            W = batched_sgd_fit(
                ds, spec, self._loss, self.epochs, self.batch_size,
                seed=self._seed(), momentum=self.momentum,
                lr_decay=getattr(self, "lr_decay", 0.0),
                adaptive=getattr(self, "adaptive", None),
            )
            fit_time = time.perf_counter() - t0
            t1 = time.perf_counter()
            scores = batched_scores_by_fold(
                ds, W, np.asarray(model_folds),
                np.asarray(col_class, dtype=np.int32),
                n_classes=n_classes, metric=metric,
            )
            score_time = time.perf_counter() - t1
            per = fit_time / max(model_idx, 1)
            per_s = score_time / max(model_idx, 1)
            for task_id, mi, fold in local_tasks:
                results[task_id] = {
                    "task_id": task_id,
                    "test_scores": {"score": float(scores[mi])},
                    "n_test": fold_test_n[fold],
                    "fit_time": per,
                    "score_time": per_s,
                }
            # full-data columns stay as ONE host weight matrix; an
            # estimator is materialized only for the candidate the search
            # asks for (refit_fn below) — the reference refits the winner
            # on the driver instead (search.py:543-550)
            ncols_per = len(cls)
            full_ids = sorted(full_models)
            cols_np = np.asarray([
                full_models[ci] * ncols_per + j
                for ci in full_ids for j in range(ncols_per)
            ])
            import torch as _torch

            Wfull = W.index_select(
                1, _torch.as_tensor(cols_np, device=ds.device)
            ).cpu().numpy()
            refit_local = {
                "ids": {ci: k for k, ci in enumerate(full_ids)},
                "W": Wfull, "per": per,
            }
        if cluster is not None:
            out = cluster.gather_task_results(results, n_cand * n_folds)
        else:
            out = [results[i] for i in range(n_cand * n_folds)]

        ncols_per = len(cls)
        proto = self

        def refit_fn(ci):
            """Materialize the fitted estimator for candidate ``ci`` from
            its full-data weight columns (owner rank broadcasts)."""
            est = None
            if refit_local and ci in refit_local["ids"]:
                k2 = refit_local["ids"][ci]
                cols = slice(k2 * ncols_per, (k2 + 1) * ncols_per)
                est = sk_clone_without_sc(proto)
                est.set_params(**candidate_params[ci])
                est._store_fitted(ds, refit_local["W"][:, cols], is_clf)
                est.n_features_in_ = ds.f
                est.fit_time_ = refit_local["per"]
            if cluster is not None and cluster.distributed:
                est = cluster.bcast_obj(
                    est, src=ci % cluster.world_size)
            return est

        return {"tasks": out, "refit_fn": refit_fn}

    def batched_eliminate(self, X, y, removals, cv_splits, scoring,
                          cluster):
        """Score every (feature-subset × fold) in ONE masked batched
        solve (DistFeatureEliminator device path; reference scores each
        subset as its own Spark task, eliminate.py:191-210).

        Per-column feature masks pin the removed features' weights to 0
        inside the solver, so column (s, fold) trains exactly the model
        on subset s.  Returns (scores [n_sets, n_folds], refit_fn) where
        ``refit_fn(s)`` materializes the fitted estimator for subset s
        from its full-data column, coef_ sliced to the kept features.
        """
        import scipy.sparse as _sp

        if _sp.issparse(X):
            raise FallbackToGeneric("sparse X not batched")
        metric = self._device_metric(scoring)
        is_clf = isinstance(self, ClassifierMixin)
        t0 = time.perf_counter()
        ds = DeviceDataset(
            X, y,
            cluster=cluster, standardize=self.standardize,
            task="cls" if is_clf else "reg",
        )
        if not ds.set_cv_partition(cv_splits):
            raise FallbackToGeneric("cv splits do not partition the data")
        n_folds = len(cv_splits)
        n_sets = len(removals)
        if is_clf:
            k = len(ds.classes_)
            n_classes = k
            cls = (
                np.array([1], dtype=np.int32) if k == 2
                else np.arange(k, dtype=np.int32)
            )
            if metric == "roc_auc" and k > 2:
                raise FallbackToGeneric(
                    "multiclass roc_auc needs sklearn's multi_class "
                    "semantics: generic path")
        else:
            n_classes = 2
            cls = np.array([-1], dtype=np.int32)
        cpm = len(cls)

        set_ids = (
            cluster.shard_indices(n_sets) if cluster is not None
            else list(range(n_sets))
        )
        fold_train_n = [len(tr) for tr, _ in cv_splits]
        col_fold, col_class, col_lr, col_l2 = [], [], [], []
        model_folds = []
        masks = []          # per MODEL (shared by its cpm columns)
        local_rows = []     # (set_id, model_idx, fold | -2)
        model_idx = 0
        for si in set_ids:
            drop = np.asarray(removals[si], dtype=np.int64)
            m = np.ones(ds.fa, dtype=np.uint8)
            m[drop] = 0      # feature j is W row j; intercept/pad stay 1
            for fold in list(range(n_folds)) + [-2]:
                n_train = (
                    fold_train_n[fold] if fold >= 0 else ds.n
                )
                lam = self._lam(n_train)
                for cc in cls:
                    col_fold.append(fold)
                    col_class.append(cc)
                    col_lr.append(self.lr)
                    col_l2.append(lam)
                masks.append(m)
                model_folds.append(fold)
                local_rows.append((si, model_idx, fold))
                model_idx += 1

        local_scores = {}
        refit_local = {}
        if model_idx:
            feat_mask = np.repeat(
                np.stack(masks, axis=1), cpm, axis=1
            )  # [fa, n_models*cpm]
            spec = ColumnSpec(
                ds.device,
                col_fold=np.asarray(col_fold, dtype=np.int32),
                col_class=np.asarray(col_class, dtype=np.int32),
                col_lr=np.asarray(col_lr, dtype=np.float32),
                col_l2=np.asarray(col_l2, dtype=np.float32),
                feat_mask=feat_mask,
            )
            W = batched_sgd_fit(
                ds, spec, self._loss, self.epochs, self.batch_size,
                seed=self._seed(), momentum=self.momentum,
                lr_decay=getattr(self, "lr_decay", 0.0),
                adaptive=getattr(self, "adaptive", None),
            )
            scores = batched_scores_by_fold(
                ds, W, np.asarray(model_folds),
                np.asarray(col_class, dtype=np.int32),
                n_classes=n_classes, metric=metric,
            )
            per = (time.perf_counter() - t0) / max(model_idx, 1)
            Wh = W.cpu().numpy()
            for si, mi, fold in local_rows:
                if fold >= 0:
                    local_scores.setdefault(si, {})[fold] = float(
                        scores[mi])
                else:
                    refit_local[si] = (
                        Wh[:, mi * cpm:(mi + 1) * cpm], per)

        # all-gather the per-set fold scores
        if cluster is not None and cluster.distributed:
            import torch.distributed as dist

            boxes = [None] * cluster.world_size
            dist.all_gather_object(boxes, local_scores)
            merged = {}
            for b in boxes:
                merged.update(b)
        else:
            merged = local_scores
        out = np.zeros((n_sets, n_folds))
        for si, folds in merged.items():
            for fold, sc_v in folds.items():
                out[si, fold] = sc_v

        proto = self

        def refit_fn(si):
            est = None
            if si in refit_local:
                Wcols, per = refit_local[si]
                est = sk_clone_without_sc(proto)
                est._store_fitted(ds, Wcols, is_clf)
                keep = np.delete(np.arange(ds.f),
                                 np.asarray(removals[si], dtype=np.int64))
                if is_clf:
                    est.coef_ = est.coef_[:, keep]
                else:
                    est.coef_ = est.coef_[keep]
                est.n_features_in_ = len(keep)
                est.fit_time_ = per
            if cluster is not None and cluster.distributed:
                est = cluster.bcast_obj(est, src=si % cluster.world_size)
            return est

        return out, refit_fn

    def batched_multiclass_fit(self, X, y, cluster, mode="ovr"):
        """Train every one-vs-rest class (or one-vs-one pair) binary
        problem as one batched device solve; used by
        DistOneVsRest/OneVsOneClassifier.  Returns (classes, estimators)
        with estimators[i] a fitted BINARY copy of self (classes_=[0,1],
        positive class = the OvR class / the pair's second class).
        """
        if getattr(self, "class_weight", None) is not None:
            raise FallbackToGeneric(
                "class_weight is per-binary-problem: generic path"
            )
        t0 = time.perf_counter()
        ds = _make_dataset(
            X, y,
            cluster=cluster, standardize=self.standardize,
        )
        ds.set_cv_partition([])
        classes = ds.classes_
        if classes is None:
            raise FallbackToGeneric("targets look continuous")
        k = len(classes)
        if mode == "ovr":
            problems = [(i, -1) for i in range(k)]
        else:
            problems = [(j, i) for i, j in combinations(range(k), 2)]
        ids = (
            cluster.shard_indices(len(problems)) if cluster is not None
            else list(range(len(problems)))
        )
        counts = np.bincount(ds.y_int.cpu().numpy(), minlength=k)
        col_class, col_class2, col_lr, col_l2 = [], [], [], []
        for pi in ids:
            tgt, other = problems[pi]
            n_train = ds.n if other < 0 else counts[tgt] + counts[other]
            col_class.append(tgt)
            col_class2.append(other)
            col_lr.append(self.lr)
            col_l2.append(self._lam(n_train))
        local = {}
        if ids:
            spec = ColumnSpec(
                ds.device,
                col_fold=np.full(len(ids), -2, dtype=np.int32),
                col_class=np.asarray(col_class, dtype=np.int32),
                col_lr=np.asarray(col_lr, dtype=np.float32),
                col_l2=np.asarray(col_l2, dtype=np.float32),
                col_class2=np.asarray(col_class2, dtype=np.int32),
            )
            W = batched_sgd_fit(
                ds, spec, self._loss, self.epochs, self.batch_size,
                seed=self._seed(), momentum=self.momentum,
                lr_decay=getattr(self, "lr_decay", 0.0),
                adaptive=getattr(self, "adaptive", None),
            )
            per = (time.perf_counter() - t0) / len(ids)
            for ci, pi in enumerate(ids):
                est = sk_clone_without_sc(self)
                est._store_fitted(ds, W[:, ci : ci + 1], True)
                est.classes_ = np.array([0, 1])
                est.n_features_in_ = ds.f
                est.fit_time_ = per
                local[pi] = est
        if cluster is not None:
            ests = cluster.gather_task_results(local, len(problems))
        else:
            ests = [local[i] for i in range(len(problems))]
        return classes, ests

    # ------------------------------------------------------------------ #
    # device inference hook (DistPredictor): fitted linear models score
    # through one device GEMM (dense X) or the sparse forward kernel
    # (CSR X — hashed-text serving at 2^20 features without host matvec)
    # ------------------------------------------------------------------ #
    def _device_predict_fn(self, method, device):
        import scipy.sparse as _sp

        is_clf = isinstance(self, ClassifierMixin)
        if method == "predict_proba" and not hasattr(self, "predict_proba"):
            return None
        if method not in ("predict", "predict_proba",
                          "decision_function"):
            return None
        import torch

        from ..ops import hip_available

        if not hip_available():
            return None
        dev = torch.device(device)
        state = {}

        def ensure():
            if state:
                return
            coef = np.atleast_2d(np.asarray(self.coef_, dtype=np.float32))
            ncols = coef.shape[0]
            cp = (ncols + 63) // 64 * 64
            W = torch.zeros(coef.shape[1], cp, dtype=torch.float32,
                            device=dev)
            W[:, :ncols] = torch.as_tensor(
                np.ascontiguousarray(coef.T), device=dev)
            Wb = torch.zeros(cp, dtype=torch.float32, device=dev)
            Wb[:ncols] = torch.as_tensor(
                np.atleast_1d(np.asarray(self.intercept_,
                                         dtype=np.float32)), device=dev)
            state.update(W=W.contiguous(), Wb=Wb.contiguous(),
                         s=torch.ones(cp, dtype=torch.float32,
                                      device=dev), ncols=ncols)

        def scores_for(X):
            ensure()
            ncols = state["ncols"]
            if _sp.issparse(X):
                from ..ops import require_hip

                Xc = X.tocsr()
                crow = torch.as_tensor(
                    np.ascontiguousarray(Xc.indptr, dtype=np.int64),
                    device=dev)
                cidx = torch.as_tensor(
                    np.ascontiguousarray(Xc.indices, dtype=np.int32),
                    device=dev)
                cval = torch.as_tensor(
                    np.ascontiguousarray(Xc.data, dtype=np.float32),
                    device=dev)
                rows = torch.arange(Xc.shape[0], dtype=torch.int64,
                                    device=dev)
                Z = torch.empty(Xc.shape[0], state["W"].shape[1],
                                dtype=torch.float32, device=dev)
                require_hip().sp_forward(crow, cidx, cval, state["W"],
                                         state["Wb"], state["s"], rows, Z)
                return Z[:, :ncols].cpu().numpy()
            Xd = torch.as_tensor(
                np.ascontiguousarray(X, dtype=np.float32), device=dev)
            Z = Xd @ state["W"][:, :ncols] + state["Wb"][:ncols]
            return Z.cpu().numpy()

        def fn(X):
            z = scores_for(X)
            scores = z[:, 0] if z.shape[1] == 1 else z
            if method == "decision_function":
                return scores
            if method == "predict":
                if not is_clf:
                    return scores
                if scores.ndim == 1:
                    return self.classes_[(scores >= 0).astype(int)]
                return self.classes_[scores.argmax(axis=1)]
            if scores.ndim == 1:
                p = 1.0 / (1.0 + np.exp(-scores))
                return np.column_stack([1.0 - p, p])
            p = 1.0 / (1.0 + np.exp(-scores))
            p /= p.sum(axis=1, keepdims=True)
            return p

        return fn

    def _check_balanced_foldable(self, y, cv_splits):
        """class_weight='balanced' is fold-dependent in sklearn; the
        batched solve shares one row-weight plane, so it is only kept
        for stratified folds: every training fold's class counts must be
        within count-rounding (±1, +1% slack — what StratifiedKFold
        guarantees) of proportional to the full data's.  Anything else
        raises FallbackToGeneric (the generic per-task path recomputes
        'balanced' per fold, matching sklearn exactly)."""
        if getattr(self, "class_weight", None) != "balanced":
            return
        yv = np.asarray(y)
        classes, full = np.unique(yv, return_counts=True)
        n = len(yv)
        for tr, _ in cv_splits:
            cnt = np.array(
                [(yv[tr] == c).sum() for c in classes], dtype=np.float64
            )
            if (cnt == 0).any():
                raise FallbackToGeneric(
                    "class_weight='balanced': a training fold misses a "
                    "class; per-fold weights diverge — generic path"
                )
            expected = full * (len(tr) / n)
            if np.max(np.abs(cnt - expected) - 0.01 * expected) > 1.0:
                raise FallbackToGeneric(
                    "class_weight='balanced' with non-stratified folds: "
                    "per-fold weights diverge from full-data weights — "
                    "generic path (exact sklearn semantics)"
                )

    def _device_metric(self, scoring):
        if scoring is None:
            metric = (
                "accuracy" if isinstance(self, ClassifierMixin) else "r2"
            )
        elif isinstance(scoring, str):
            metric = scoring
        else:
            raise FallbackToGeneric("callable/multi scoring not batched")
        if metric not in _DEVICE_METRICS:
            raise FallbackToGeneric(f"no device metric for {metric!r}")
        return metric


def _lam_from(params, n_train):
    if "C" in params and params.get("C") is not None:
        return 1.0 / (float(params["C"]) * max(n_train, 1))
    return float(params.get("alpha", 1.0)) / max(n_train, 1)


def sk_clone_without_sc(est):
    """Unfitted copy of est with sc=None (for batch-materialized models)."""
    from sklearn.base import clone

    sc = getattr(est, "sc", None)
    est.sc = None
    try:
        out = clone(est)
    finally:
        est.sc = sc
    return out


class LogisticRegression(ClassifierMixin, _BatchedLinearBase):
    """Batched-SGD logistic regression (binary + internal one-vs-rest).

    The solver differs from sklearn's lbfgs/liblinear (SURVEY.md §7 "exact
    sklearn numerics"): features are standardized internally (coefficients
    are mapped back to raw space, so the fitted model is exchangeable),
    and the optimizer is mini-batch SGD — scores match sklearn to CV-noise
    tolerance, not bitwise.

    Wide/huge sparse X trains on the sparse-native solver
    (models/_sparse_sgd.py), which by design skips standardization
    (centering would densify; hashed text is already unit-scale), runs
    momentum-free, and Adagrad-normalizes the data-gradient steps
    unless ``adaptive=False``.
    """

    _loss = LOSS_LOG

    def __init__(self, C=1.0, lr=0.5, epochs=20, batch_size=8192,
                 momentum=0.9, lr_decay=0.0, standardize=True,
                 class_weight=None, random_state=None, adaptive=None,
                 sc=None):
        self.C = C
        self.adaptive = adaptive
        self.class_weight = class_weight
        self.lr_decay = lr_decay
        self.lr = lr
        self.epochs = epochs
        self.batch_size = batch_size
        self.momentum = momentum
        self.standardize = standardize
        self.random_state = random_state
        self.sc = sc

    def predict_proba(self, X):
        scores = self.decision_function(X)
        if scores.ndim == 1:
            p = 1.0 / (1.0 + np.exp(-scores))
            return np.column_stack([1.0 - p, p])
        p = 1.0 / (1.0 + np.exp(-scores))
        p /= p.sum(axis=1, keepdims=True)
        return p

    def predict_log_proba(self, X):
        return np.log(np.clip(self.predict_proba(X), 1e-300, None))

    def predict(self, X):
        scores = self.decision_function(X)
        if scores.ndim == 1:
            return self.classes_[(scores >= 0).astype(int)]
        return self.classes_[scores.argmax(axis=1)]


class LinearSVC(ClassifierMixin, _BatchedLinearBase):
    """Batched-SGD linear SVM (hinge loss, internal one-vs-rest)."""

    _loss = LOSS_HINGE

    def __init__(self, C=1.0, lr=0.5, epochs=20, batch_size=8192,
                 momentum=0.9, lr_decay=0.0, standardize=True,
                 class_weight=None, random_state=None, adaptive=None,
                 sc=None):
        self.C = C
        self.adaptive = adaptive
        self.class_weight = class_weight
        self.lr_decay = lr_decay
        self.lr = lr
        self.epochs = epochs
        self.batch_size = batch_size
        self.momentum = momentum
        self.standardize = standardize
        self.random_state = random_state
        self.sc = sc

    def predict(self, X):
        scores = self.decision_function(X)
        if scores.ndim == 1:
            return self.classes_[(scores >= 0).astype(int)]
        return self.classes_[scores.argmax(axis=1)]


class Ridge(RegressorMixin, _BatchedLinearBase):
    """Batched-SGD ridge regression.

    Default ``momentum=0.0``: unlike the bounded-gradient losses
    (logistic/hinge, which keep 0.9), heavy momentum at these learning
    rates oscillates on squared loss and slows convergence several-fold
    (measured: CV r2 0.89 vs 0.998 at 10 epochs on 12k x 16 synthetic).
    """

    _loss = LOSS_SQUARED

    def __init__(self, alpha=1.0, lr=0.5, epochs=20, batch_size=8192,
                 momentum=0.0, lr_decay=0.0, standardize=True,
                 random_state=None, adaptive=None, sc=None):
        self.alpha = alpha
        self.adaptive = adaptive
        self.lr_decay = lr_decay
        self.lr = lr
        self.epochs = epochs
        self.batch_size = batch_size
        self.momentum = momentum
        self.standardize = standardize
        self.random_state = random_state
        self.sc = sc

    def predict(self, X):
        return self.decision_function(X)
