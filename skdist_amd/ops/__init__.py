"""
HIP/CDNA4 kernel extension loader.

The extension (`_skdist_hip`) is built IN-TREE for gfx950 by
``__graft_entry__.build()`` / ``python setup.py build_ext --inplace`` and
ships with the repo snapshot.  Policy on a GPU box: the HIP kernels ARE the
compute path — if the extension is missing, ops raise instead of silently
falling back to eager torch (set SKDIST_AMD_ALLOW_EAGER=1 to waive, for
debugging only).
"""

import os

import torch

_ext = None
_ext_err = None


def _load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from . import _skdist_hip  # built in-tree, travels with the repo

        _ext = _skdist_hip
    except ImportError as e:
        _ext_err = e
    return _ext


def hip_available():
    return torch.cuda.is_available() and _load() is not None


def require_hip():
    if not torch.cuda.is_available():
        raise RuntimeError("require_hip(): no HIP device visible")
    if _load() is None:
        raise RuntimeError(
            "skdist_amd HIP extension is not built — run "
            "`python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"(import error: {_ext_err})"
        )
    return _ext


def sgd_step_hip(Xaug, y_float, fold_id, idx, W, V, spec, loss_id,
                 lr_scale, momentum):
    """One fused mini-batch SGD step on the HIP kernels."""
    ext = require_hip()
    ext.sgd_step(
        Xaug, y_float, fold_id, idx, W,
        V if V is not None else torch.empty(0, device=W.device),
        spec.col_fold, spec.col_class, spec.col_lr, spec.col_l2,
        int(loss_id), float(lr_scale), float(momentum),
    )
