import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)  # for oracle/ package
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))  # caffe_amd


def _has_gpu():
    # deliberately NOT via torch: importing torch loads its bundled ROCm
    # runtime into the process, which shadows /opt/rocm's and breaks the
    # engine's device detection (observed on the GPU box).  The engine and
    # its tests never need torch; only multi-process gloo tests import it,
    # in CPU mode.
    return os.path.exists("/dev/kfd")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    has_gpu = _has_gpu()
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(skip)
