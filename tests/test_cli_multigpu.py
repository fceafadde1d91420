"""`caffe train -gpu=all` — the reference's multi-GPU command line
(tools/caffe.cpp:154 + P2PManager, parallel.cpp).  Here the engine forks
one worker process per visible device (MI355X-native one-process-per-GPU
instead of the reference's thread-per-GPU), rendezvous over a tmp file,
and the parent prints the reference's overall line (parallel.cpp:85).

On the 1-GPU CI box this exercises the scout/fork/perf-aggregation path
with world 1; the 8-GPU run is the driver's scaling bench (bench.py).
"""
import os
import subprocess
import sys
import tempfile

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CAFFE = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from test_cli_snapshot import make_lenet_solver  # noqa: E402

pytestmark = pytest.mark.gpu


def test_train_gpu_all_forks_and_reports():
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_lenet_solver(tmp, extra="test_interval: 0")
        env = dict(os.environ, CAFFE_SYN_SHAPE="1x28x28x10")
        out = subprocess.run(
            [CAFFE, "train", f"-solver={solver}", "-gpu=all",
             "-iterations=6"],
            capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
        assert out.returncode == 0, out.stderr
        assert "Optimization Done." in out.stderr, out.stderr
        assert "Overall multi-GPU performance:" in out.stderr, out.stderr
        # rank 0 snapshots
        assert any(f.endswith(".caffemodel") for f in os.listdir(tmp)), \
            out.stderr
