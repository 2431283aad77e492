"""
In-tree build of the gfx950 HIP extension.

Drives hipcc directly (no hipify, no JIT cache): the built
``_skdist_hip.so`` lands next to this file so it travels with the repo
snapshot to the GPU box.  Usage:

    python -m skdist_amd.ops.build          # or __graft_entry__.build()
"""

import os
import subprocess
import sys
import sysconfig

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT = os.path.join(HERE, "_skdist_hip.so")

KERNEL_SOURCES = ["sgd_kernels.hip", "sparse_sgd_kernels.hip",
                  "tree_kernels.hip", "predict_kernels.hip",
                  "hash_kernels.hip"]
BINDING_SOURCES = ["bindings.cpp"]


def _torch_flags():
    import torch
    from torch.utils import cpp_extension as ce

    includes = [f"-I{p}" for p in ce.include_paths()]
    includes.append(f"-I{sysconfig.get_paths()['include']}")
    lib_dirs = [f"-L{p}" for p in ce.library_paths()]
    libs = ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10"]
    if os.path.exists(
        os.path.join(ce.library_paths()[0], "libtorch_hip.so")
    ):
        libs += ["-ltorch_hip", "-lc10_hip"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defines = [
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_skdist_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
    ]
    return includes, lib_dirs, libs, defines


def _run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build(verbose=True):
    includes, lib_dirs, libs, defines = _torch_flags()
    objs = []
    common = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-Wall", "-Wno-unused-function",
    ]
    for src in KERNEL_SOURCES:
        path = os.path.join(CSRC, src)
        if not os.path.exists(path):
            continue
        obj = os.path.join(CSRC, src.rsplit(".", 1)[0] + ".o")
        if not _stale(obj, [path, os.path.join(CSRC, "common.h")]):
            objs.append(obj)
            continue
        _run(common + ["-c", path, "-o", obj])
        objs.append(obj)
    for src in BINDING_SOURCES:
        path = os.path.join(CSRC, src)
        obj = os.path.join(CSRC, src.rsplit(".", 1)[0] + ".o")
        if _stale(obj, [path]):
            _run(common + includes + defines + ["-fno-gpu-rdc", "-c", path,
                                                "-o", obj])
        objs.append(obj)
    _run(common + ["-shared", "-o", OUT] + objs + lib_dirs + libs)
    print(f"built {OUT}")
    return OUT


def _stale(obj, deps):
    if not os.path.exists(obj):
        return True
    omt = os.path.getmtime(obj)
    return any(os.path.getmtime(d) > omt for d in deps)


def extension_is_stale():
    """True when any csrc file is newer than the built .so (a stale .so
    on the GPU box runs OLD kernels — fail loudly instead)."""
    if not os.path.exists(OUT):
        return True
    omt = os.path.getmtime(OUT)
    for fn in os.listdir(CSRC):
        if fn.endswith((".hip", ".cpp", ".h")):
            if os.path.getmtime(os.path.join(CSRC, fn)) > omt:
                return True
    return False


if __name__ == "__main__":
    build()
    sys.exit(0)
