"""
Sparse-native batched SGD linear solver — the text-scale path.

The dense solver (``_sgd.py``) batches every (candidate x fold x class)
model into columns of one MFMA GEMM against a dense HBM-resident X; for
hashed-text features (sk-dist's flagship real workloads: reference
``_defaults.py:91-198``, ``examples/postprocessing/simple_voter.py:23-42``)
a 1M x 2^20 matrix densifies to ~4 TB, so this module keeps X as device
CSR and trains the same column batch with gather/scatter kernels
(``ops/csrc/sparse_sgd_kernels.hip``):

  * forward: one lane owns (row, column); a feature's weights for 64
    consecutive columns are one coalesced 256 B gather of the row-major
    fp32 table ``W [f][CP]``;
  * loss gradient fused into the forward epilogue (same dloss / fold
    mask / one-vs-one pair mask / row weights as the dense ``k_fwd_gt``),
    staged as bf16 ``G [m][CP]``;
  * backward: a per-batch CSC (built ONCE on device, shared by every
    epoch since minibatches are fixed slabs of one seeded shuffle — the
    dense path's contract) lets one lane own (feature, column): no
    atomics, deterministic;
  * L2 decay of all f rows is applied lazily as a per-column scale
    (exactly the dense update's math; see the kernel header).

Solver deltas vs the dense path, by design: no standardization (hashed
text is already unit-scale; centering would densify) and no momentum
(a momentum buffer adds an O(f x C) touch per step for features the
batch never saw; convex SGD on text converges fine without it — the
driver warns once when a nonzero momentum is dropped).

The eager torch mirror (`sparse_sgd_fit(..., force_eager=True)` /
CPU path) replicates the kernel's number flow (fp32 accumulate, bf16 G)
and is the numerics reference the GPU tests compare against.
"""

import os
import warnings

import numpy as np
import torch

from ._sgd import _LOSS_IDS, _MetricState, _use_hip, encode_labels

LOSS_LOG, LOSS_HINGE, LOSS_SQUARED = 0, 1, 2


def _pad_cols(c):
    """Column padding rule shared with the kernels (multiple of 64; the
    kernels bounds-guard the tail column tile)."""
    return (c + 63) // 64 * 64


class SparseDeviceDataset:
    """CSR (X, y) resident on one device, shared by every fit in a
    search — the sparse twin of ``_sgd.DeviceDataset`` (same label
    normalization, fold encoding and cluster broadcast contract).
    """

    is_sparse = True

    def __init__(self, X, y, cluster=None, device=None,
                 sample_weight=None, task=None, classes=None):
        import scipy.sparse as sp

        self.cluster = cluster
        if device is None:
            device = cluster.device if cluster is not None else (
                torch.device("cuda") if torch.cuda.is_available()
                else torch.device("cpu")
            )
        self.device = torch.device(device)
        self.comp_dtype = torch.float32  # master weights; G is bf16

        # host-side label normalization (same rules as DeviceDataset)
        kind = cls_arr = y_host = None
        if y is not None:
            y_np = np.asarray(y)
            if classes is not None:
                cls_arr = np.asarray(classes)
                y_host = np.ascontiguousarray(
                    np.searchsorted(cls_arr, y_np), dtype=np.int32)
                kind = "cls"
            elif task == "reg" or (task is None and y_np.dtype.kind == "f"):
                y_host = np.ascontiguousarray(y_np, dtype=np.float32)
                kind = "reg"
            else:
                cls_arr, enc = encode_labels(y_np)
                y_host = np.ascontiguousarray(enc, dtype=np.int32)
                kind = "cls"

        crow = cidx = cval = None
        if X is not None:
            Xc = sp.csr_matrix(X) if not sp.isspmatrix_csr(X) else X
            Xc.sort_indices()
            crow = torch.as_tensor(
                np.ascontiguousarray(Xc.indptr, dtype=np.int64))
            cidx = torch.as_tensor(
                np.ascontiguousarray(Xc.indices, dtype=np.int32))
            cval = torch.as_tensor(
                np.ascontiguousarray(Xc.data, dtype=np.float32))
            self._shape = Xc.shape

        if cluster is not None and cluster.distributed:
            import torch.distributed as dist

            dev = cluster.device
            # SPMD fast path (see DeviceDataset): all ranks holding the
            # identical host CSR upload locally, skipping the broadcast
            have = [None] * cluster.world_size
            dist.all_gather_object(
                have, crow is not None and y_host is not None)
            if all(have):
                crow = crow.to(dev)
                cidx = cidx.to(dev)
                cval = cval.to(dev)
                yt = torch.as_tensor(y_host).to(dev)
            else:
                crow = cluster.bcast_tensor(
                    crow.to(dev) if crow is not None else None)
                cidx = cluster.bcast_tensor(
                    cidx.to(dev) if cidx is not None else None)
                cval = cluster.bcast_tensor(
                    cval.to(dev) if cval is not None else None)
                yt = cluster.bcast_tensor(
                    torch.as_tensor(y_host).to(dev)
                    if y_host is not None else None)
            kind, cls_arr, shape = cluster.bcast_obj(
                (kind, cls_arr, getattr(self, "_shape", None)))
            if shape is not None:
                self._shape = shape
        else:
            crow = crow.to(self.device)
            cidx = cidx.to(self.device)
            cval = cval.to(self.device)
            yt = torch.as_tensor(y_host).to(self.device)

        from ._sgd import check_finite

        check_finite(cval, "X (sparse values)")
        check_finite(yt, "y")
        self.crow, self.cidx, self.cval = crow, cidx, cval
        self.n = crow.shape[0] - 1
        self.f = (
            self._shape[1] if hasattr(self, "_shape")
            else int(cidx.max().item()) + 1
        )
        self.fa = self.f + 1
        self.intercept_row = self.f
        if kind == "reg":
            self.classes_ = None
            self.y_float = yt.to(torch.float32)
            self.y_int = None
        else:
            self.classes_ = cls_arr
            self.y_int = yt.to(torch.int32)
            self.y_float = self.y_int.to(torch.float32)

        if sample_weight is not None:
            self.row_w = torch.as_tensor(
                np.ascontiguousarray(sample_weight, dtype=np.float32),
                device=self.device)
        else:
            self.row_w = None
        self.fold_id = None
        self._shuf_key = None

    # identical contract to DeviceDataset.set_cv_partition
    def set_cv_partition(self, cv_splits):
        if not cv_splits:
            self.fold_id = torch.full(
                (self.n,), -1, dtype=torch.int32, device=self.device)
            return True
        fold = np.full(self.n, -1, dtype=np.int32)
        for k, (_, test_idx) in enumerate(cv_splits):
            if np.any(fold[test_idx] != -1):
                return False
            fold[test_idx] = k
        if np.any(fold == -1):
            return False
        self.fold_id = torch.as_tensor(fold, device=self.device)
        return True

    def unstandardize_coef(self, w, b):
        return w, b  # sparse path never standardizes

    # ------------------------------------------------------------------ #
    def shuffled(self, seed, batch_size):
        """Shuffled CSR + per-batch CSC, cached (every solve in a search
        shares the seed and batch size, so the sort/segment work is paid
        once).  Returns a dict of device tensors + CPU metadata."""
        key = (int(seed), int(batch_size))
        if self._shuf_key == key:
            return self._shuf
        dev = self.device
        n, f = self.n, self.f
        rng = np.random.default_rng(seed)
        permt = torch.as_tensor(rng.permutation(n), dtype=torch.int64,
                                device=dev)
        lens = self.crow[1:] - self.crow[:-1]
        slens = lens.index_select(0, permt)
        scrow = torch.zeros(n + 1, dtype=torch.int64, device=dev)
        torch.cumsum(slens, 0, out=scrow[1:])
        nnz = int(self.crow[-1].item())
        row_of_pos = torch.repeat_interleave(
            torch.arange(n, dtype=torch.int64, device=dev), slens)
        pos_in_row = (
            torch.arange(nnz, dtype=torch.int64, device=dev)
            - scrow[:-1].index_select(0, row_of_pos)
        )
        src_start = self.crow[:-1].index_select(0, permt)
        gather = src_start.index_select(0, row_of_pos) + pos_in_row
        scidx = self.cidx.index_select(0, gather).contiguous()
        scval = self.cval.index_select(0, gather).contiguous()
        ys = self.y_float.index_select(0, permt).contiguous()
        folds = self.fold_id.index_select(0, permt).contiguous()
        rw = (
            self.row_w.index_select(0, permt).contiguous()
            if self.row_w is not None else None
        )

        # batch CSC: stable sort by (batch, feature); rows within a
        # group keep CSR order, so ridx is sorted per (batch, feature)
        batch = row_of_pos // batch_size
        skey = batch * f + scidx.to(torch.int64)
        order = torch.argsort(skey, stable=True)
        skey_s = skey.index_select(0, order)
        ridx = (row_of_pos % batch_size).index_select(0, order).to(
            torch.int32).contiguous()
        bval = scval.index_select(0, order).contiguous()
        uniq, counts = torch.unique_consecutive(skey_s,
                                                return_counts=True)
        cptr = torch.zeros(len(uniq) + 1, dtype=torch.int64, device=dev)
        torch.cumsum(counts, 0, out=cptr[1:])
        ufeat = (uniq % f).to(torch.int32).contiguous()
        ubatch = uniq // f
        n_batches = (n + batch_size - 1) // batch_size
        ub_ptr = torch.searchsorted(
            ubatch, torch.arange(n_batches + 1, dtype=torch.int64,
                                 device=dev)).cpu()
        if rw is None:
            inv_m = torch.tensor(
                [1.0 / min(batch_size, n - s0)
                 for s0 in range(0, n, batch_size)], dtype=torch.float32)
        else:
            sums = [
                float(rw[s0: s0 + batch_size].sum())
                for s0 in range(0, n, batch_size)
            ]
            inv_m = torch.tensor(
                [1.0 / max(v, 1e-30) for v in sums], dtype=torch.float32)
        self._shuf = {
            "crow": scrow.contiguous(), "cidx": scidx, "cval": scval,
            "y": ys, "fold": folds, "rw": rw,
            "ufeat": ufeat, "cptr": cptr, "ridx": ridx, "bval": bval,
            "ub_ptr": ub_ptr, "inv_m": inv_m, "perm": permt,
        }
        self._shuf_key = key
        return self._shuf


# --------------------------------------------------------------------- #
# the solve
# --------------------------------------------------------------------- #

def sparse_sgd_fit(ds, spec, loss, epochs, batch_size, seed=0,
                   momentum=0.0, lr_decay=0.0, force_eager=False,
                   adaptive=None):
    """Train all columns on the sparse dataset; returns W [f+1, ncols]
    fp32 on ds.device (row f = intercept), same contract as
    ``batched_sgd_fit``.  ``momentum`` > 0 is dropped with a one-time
    warning (module docstring).

    ``adaptive`` (default True): Adagrad-normalize the data-gradient
    step per (feature, column) — rare text features take full-size
    first steps instead of 1/batch-size ones.  The lazy L2 decay is
    unaffected.  ``adaptive=False`` runs the dense solver's plain-SGD
    update (the dense-equivalence mode the tests use)."""
    if momentum > 0.0:
        warnings.warn(
            "sparse batched solve runs momentum-free SGD (a momentum "
            "buffer would touch all f weight rows every step); set "
            "momentum=0 to silence", stacklevel=2)
    if getattr(spec, "feat_mask", None) is not None:
        raise ValueError("feat_mask is not supported on the sparse path")
    if adaptive is None:
        adaptive = True
    loss_id = _LOSS_IDS[loss] if isinstance(loss, str) else loss
    dev = ds.device
    ncols = spec.ncols
    cp = _pad_cols(ncols)
    sh = ds.shuffled(seed, batch_size)

    col = _padded_cols(spec, cp, dev)
    # W rows 0..f-1 are the kernels' scaled table; row f is plain
    # storage for the intercept (kernels never touch it: cidx < f)
    W = torch.zeros(ds.f + 1, cp, dtype=torch.float32, device=dev)
    Wb = torch.zeros(cp, dtype=torch.float32, device=dev)
    s = torch.ones(cp, dtype=torch.float32, device=dev)
    if adaptive:
        H = torch.zeros(ds.f, cp, dtype=torch.float32, device=dev)
        Hb = torch.zeros(cp, dtype=torch.float32, device=dev)
    else:
        H = torch.empty(0, dtype=torch.float32, device=dev)
        Hb = torch.empty(0, dtype=torch.float32, device=dev)

    hip = (not force_eager) and _use_hip(dev)
    if hip:
        from ..ops import require_hip

        ext = require_hip()
        m0 = min(batch_size, ds.n)
        G = torch.empty(m0, cp, dtype=torch.bfloat16, device=dev)
        # k_sp_colsum chunking caps n_chunks at 256 for any batch size
        part = torch.empty(256, cp, dtype=torch.float32, device=dev)
        rw = (
            sh["rw"] if sh["rw"] is not None
            else torch.empty(0, dtype=torch.float32, device=dev)
        )
        # host-side fp32 mirror of the lazy-scale decay (same sequential
        # multiplies as k_sp_bias_scale), so the renorm check never
        # reads the device — epochs enqueue back-to-back with no syncs
        s_host = np.ones(cp, dtype=np.float32)
        lr_h = spec.col_lr.cpu().numpy().astype(np.float32)
        l2_h = spec.col_l2.cpu().numpy().astype(np.float32)
        lr_pad = np.zeros(cp, dtype=np.float32)
        l2_pad = np.zeros(cp, dtype=np.float32)
        lr_pad[: len(lr_h)] = lr_h
        l2_pad[: len(l2_h)] = l2_h
        n_batches = (ds.n + batch_size - 1) // batch_size
        # precompute whether the renorm ever fires (the scale decay is
        # fully deterministic): if it never does and lr is constant,
        # all epochs go in ONE call — epochs 2..N replay a hipGraph
        renorm_free = lr_decay == 0.0
        if renorm_free:
            f0 = np.float32(1.0) - lr_pad * l2_pad
            fmin = float(f0.min())
            if fmin <= 0.0:
                renorm_free = False
            elif fmin < 1.0:
                # closed form with a 2x margin over the 1e-3 runtime
                # threshold (covers fp32 sequential-multiply drift)
                renorm_free = (
                    epochs * n_batches * np.log(fmin) > np.log(2e-3)
                )
        if renorm_free:
            ext.sp_sgd_solve(
                sh["crow"], sh["cidx"], sh["cval"], W[: ds.f], Wb, s,
                H, Hb, G, part, sh["y"], sh["fold"], rw,
                col["cls"], col["fold"], col["cls2"], col["lr"],
                col["l2"], sh["ufeat"], sh["cptr"], sh["ridx"],
                sh["bval"], sh["ub_ptr"], sh["inv_m"],
                int(batch_size), int(loss_id), int(epochs))
        else:
            for epoch in range(epochs):
                lr_scale = 1.0 / (1.0 + lr_decay * epoch)
                ext.sp_sgd_epoch(
                    sh["crow"], sh["cidx"], sh["cval"], W[: ds.f], Wb, s,
                    H, Hb, G, part, sh["y"], sh["fold"], rw,
                    col["cls"], col["fold"], col["cls2"], col["lr"],
                    col["l2"], sh["ufeat"], sh["cptr"], sh["ridx"],
                    sh["bval"], sh["ub_ptr"], sh["inv_m"],
                    int(batch_size), int(loss_id), float(lr_scale))
                f = (np.float32(1.0)
                     - (lr_pad * np.float32(lr_scale)) * l2_pad)
                for _ in range(n_batches):
                    s_host *= f
                if float(s_host.min()) < 1e-3:
                    ext.sp_renorm(W[: ds.f], s)
                    s.fill_(1.0)
                    s_host[:] = 1.0
    else:
        _sparse_sgd_eager(ds, sh, W, Wb, s, H, Hb, col, loss_id, epochs,
                          batch_size, lr_decay, cp)

    W[: ds.f].mul_(s.unsqueeze(0))
    W[ds.f] = Wb
    return W[:, :ncols]


def _padded_cols(spec, cp, dev):
    def pad(t, fill, dtype):
        out = torch.full((cp,), fill, dtype=dtype, device=dev)
        out[: t.shape[0]] = t
        return out.contiguous()

    return {
        "cls": pad(spec.col_class, -99, torch.int32),
        "fold": pad(spec.col_fold, -9, torch.int32),
        "cls2": pad(spec.col_class2, -1, torch.int32),
        "lr": pad(spec.col_lr, 0.0, torch.float32),
        "l2": pad(spec.col_l2, 0.0, torch.float32),
    }


def _sparse_sgd_eager(ds, sh, W, Wb, s, H, Hb, col, loss_id, epochs,
                      batch_size, lr_decay, cp):
    """Eager torch mirror of the kernels' number flow (fp32 math, bf16
    G round-trip, lazy L2 scale).  The numerics reference for the GPU
    tests; also the CPU execution path."""
    dev = ds.device
    n, f = ds.n, ds.f
    crow, cidx, cval = sh["crow"], sh["cidx"], sh["cval"]
    y, fold, rw = sh["y"], sh["fold"], sh["rw"]
    inv_m = sh["inv_m"]
    Wt = W[:f]
    for epoch in range(epochs):
        lr_scale = 1.0 / (1.0 + lr_decay * epoch)
        for bi, start in enumerate(range(0, n, batch_size)):
            m = min(batch_size, n - start)
            k0 = int(crow[start])
            k1 = int(crow[start + m])
            jj = cidx[k0:k1].to(torch.int64)
            vv = cval[k0:k1]
            rl = (
                torch.repeat_interleave(
                    torch.arange(m, dtype=torch.int64, device=dev),
                    crow[start + 1: start + m + 1]
                    - crow[start: start + m])
            )
            contrib = vv.unsqueeze(1) * Wt.index_select(0, jj)
            Z = torch.zeros(m, cp, dtype=torch.float32, device=dev)
            Z.index_add_(0, rl, contrib)
            Z = s.unsqueeze(0) * Z + Wb.unsqueeze(0)

            yb = y[start: start + m]
            cls = col["cls"].to(torch.float32)
            t = (yb.unsqueeze(1) == cls.unsqueeze(0)).to(torch.float32)
            if loss_id == LOSS_SQUARED:
                reg = col["cls"] < 0
                t[:, reg] = yb.unsqueeze(1).expand(m, int(reg.sum()))
            if loss_id == LOSS_LOG:
                G = torch.sigmoid(Z.clamp(-30, 30)) - t
            elif loss_id == LOSS_HINGE:
                sgn = 2.0 * t - 1.0
                G = torch.where(sgn * Z < 1.0, -sgn,
                                torch.zeros_like(Z))
            else:
                G = Z - t
            mask = fold[start: start + m].unsqueeze(1) != col[
                "fold"].unsqueeze(0)
            c2 = col["cls2"].unsqueeze(0)
            pair_ok = (
                (c2 < 0)
                | (yb.unsqueeze(1) == cls.unsqueeze(0))
                | (yb.unsqueeze(1) == c2.to(torch.float32))
            )
            G = G * (mask & pair_ok).to(torch.float32)
            if rw is not None:
                G = G * rw[start: start + m].unsqueeze(1)
            G = G.to(torch.bfloat16).to(torch.float32)  # kernel stores bf16

            im = float(inv_m[bi])
            lr = col["lr"] * lr_scale
            gb = G.sum(dim=0) * im
            if Hb.numel():
                Hb.add_(gb * gb)
                Wb.sub_(lr * gb * torch.rsqrt(Hb + 1e-12))
            else:
                Wb.sub_(lr * gb)
            s.mul_(1.0 - lr * col["l2"])
            GW = torch.zeros_like(Wt)
            GW.index_add_(0, jj, vv.unsqueeze(1) * G.index_select(0, rl))
            GW.mul_(im)
            if H.numel():
                H.add_(GW * GW)
                Wt.sub_(lr.unsqueeze(0) * GW * torch.rsqrt(H + 1e-12)
                        / s.unsqueeze(0))
            else:
                Wt.sub_(lr.unsqueeze(0) * GW / s.unsqueeze(0))
        # renorm check per epoch, matching the HIP driver
        if float(s.min()) < 1e-3:
            Wt.mul_(s.unsqueeze(0))
            s.fill_(1.0)


# --------------------------------------------------------------------- #
# batched fold scoring (sparse forward -> shared _MetricState)
# --------------------------------------------------------------------- #

def sparse_scores_by_fold(ds, W, model_folds, col_class, n_classes,
                          metric, chunk=1 << 17):
    """Per-model test-fold metrics for the sparse path; same contract as
    ``_sgd.batched_scores_by_fold`` (W is the assembled [f+1, ncols])."""
    dev = ds.device
    cpm = n_classes if n_classes > 2 else 1
    n_models = len(model_folds)
    out = np.zeros(n_models)
    hip = _use_hip(dev)
    if hip:
        from ..ops import require_hip

        ext = require_hip()
    ones = torch.ones(1, dtype=torch.float32, device=dev)

    class _Shim:
        pass

    for fsel in np.unique(model_folds[model_folds >= 0]):
        mids = np.flatnonzero(model_folds == fsel)
        cols_np = (mids[:, None] * cpm + np.arange(cpm)).ravel()
        nsel = len(cols_np)
        cp = _pad_cols(nsel)
        Wf = torch.zeros(ds.f + 1, cp, dtype=torch.float32, device=dev)
        Wf[:, :nsel] = W.index_select(
            1, torch.as_tensor(cols_np, device=dev))
        Wsel = Wf[: ds.f].contiguous()
        Wbsel = Wf[ds.f].contiguous()
        ssel = ones.expand(cp).contiguous()
        rows = torch.nonzero(ds.fold_id == int(fsel)).flatten()
        nm = len(mids)
        spec = _Shim()
        spec.col_class = torch.as_tensor(
            np.ascontiguousarray(col_class[cols_np]), device=dev)
        model_fold_t = torch.full((nm,), int(fsel), dtype=torch.int32,
                                  device=dev)
        state = _MetricState(metric, nm, n_classes, dev)
        for lo in range(0, len(rows), chunk):
            r = rows[lo: lo + chunk].to(torch.int64).contiguous()
            Z = torch.empty(len(r), cp, dtype=torch.float32, device=dev)
            if hip:
                ext.sp_forward(ds.crow, ds.cidx, ds.cval, Wsel, Wbsel,
                               ssel, r, Z)
            else:
                _sp_forward_eager(ds, Wsel, Wbsel, r, Z)
            fid = ds.fold_id.index_select(0, r)
            yb = ds.y_float.index_select(0, r)
            # _MetricState expects [m, nm*cpm] column blocks
            state.update(Z[:, :nsel], yb, fid, spec, model_fold_t,
                         n_classes)
        out[mids] = state.finalize()
    return out


def _sp_forward_eager(ds, Wt, Wb, rows, Z):
    """Eager sparse forward for the scoring path (CPU / reference)."""
    dev = ds.device
    m = len(rows)
    lens = (ds.crow[1:] - ds.crow[:-1]).index_select(0, rows)
    rl = torch.repeat_interleave(
        torch.arange(m, dtype=torch.int64, device=dev), lens)
    starts = ds.crow[:-1].index_select(0, rows)
    off = torch.zeros(m + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=off[1:])
    pos = torch.arange(int(off[-1]), dtype=torch.int64, device=dev) - \
        off[:-1].index_select(0, rl)
    gather = starts.index_select(0, rl) + pos
    jj = ds.cidx.index_select(0, gather).to(torch.int64)
    vv = ds.cval.index_select(0, gather)
    Z.zero_()
    Z.index_add_(0, rl, vv.unsqueeze(1) * Wt.index_select(0, jj))
    Z.add_(Wb.unsqueeze(0))
