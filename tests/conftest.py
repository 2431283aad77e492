import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a HIP device (run on the MI355X box)"
    )


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no HIP device in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def pytest_sessionstart(session):
    # a stale in-tree extension would ship OLD kernels to the GPU box;
    # fail the CPU suite early instead
    from skdist_amd.ops.build import extension_is_stale

    if extension_is_stale():
        raise RuntimeError(
            "skdist_amd/ops/_skdist_hip.so is stale or missing — run "
            "`python -m skdist_amd.ops.build`"
        )
