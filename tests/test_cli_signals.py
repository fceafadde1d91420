"""Signal surface of `caffe train` (reference tools/caffe.cpp:31-36 +
util/signal_handler.cpp): SIGINT defaults to stop (finish the iteration,
snapshot_after_train still applies), SIGHUP defaults to snapshot-and-
continue; both remappable via -sigint_effect / -sighup_effect.
"""
import glob
import os
import signal
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CAFFE = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from test_cli_snapshot import run_env  # noqa: E402

# a deliberately tiny net (one small IP) so a single iteration stays in the
# millisecond range even on a heavily loaded machine — the signal tests
# bound how long the process may take to REACT, which is measured in
# iterations
TINY_SOLVER = """base_lr: 0.01
lr_policy: "fixed"
snapshot_prefix: "{prefix}"
max_iter: 100000000
net_param {{
  name: "tiny"
  layer {{
    name: "input"
    type: "Input"
    top: "in0"
    top: "in1"
    input_param {{
      shape {{ dim: 4 dim: 8 }}
      shape {{ dim: 4 }}
    }}
  }}
  layer {{
    name: "ip"
    type: "InnerProduct"
    bottom: "in0"
    top: "fc"
    inner_product_param {{ num_output: 3 }}
  }}
  layer {{
    name: "loss"
    type: "SoftmaxWithLoss"
    bottom: "fc"
    bottom: "in1"
    top: "loss"
  }}
}}
"""


def make_tiny_solver(tmp, extra=""):
    path = os.path.join(tmp, "solver.prototxt")
    with open(path, "w") as f:
        f.write(TINY_SOLVER.format(prefix=os.path.join(tmp, "lenet")))
        f.write(extra + "\n")
    return path


def start_train(solver, extra=()):
    env = dict(run_env(), CAFFE_SYN_SHAPE="1x28x28x10")
    return subprocess.Popen(
        [CAFFE, "train", f"-solver={solver}", "-iterations=1000000000",
         *extra],
        env=env, cwd=REPO,
        stdout=subprocess.PIPE, stderr=subprocess.PIPE)


def test_sigint_stops_training():
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_tiny_solver(tmp)
        p = start_train(solver)
        time.sleep(3)  # past model build, mid-Step
        p.send_signal(signal.SIGINT)
        _, err = p.communicate(timeout=300)
        err = err.decode()
        assert p.returncode == 0, err
        assert "Optimization stopped early." in err, err
        assert "Optimization Done." in err, err
        # snapshot_after_train still applies on early exit
        assert glob.glob(os.path.join(tmp, "lenet_iter_*.caffemodel")), err


def test_sighup_snapshots_and_continues():
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_tiny_solver(tmp)
        p = start_train(solver)
        time.sleep(3)
        p.send_signal(signal.SIGHUP)
        deadline = time.time() + 60
        snaps = []
        while time.time() < deadline:
            snaps = glob.glob(os.path.join(tmp, "lenet_iter_*.solverstate"))
            if snaps:
                break
            assert p.poll() is None, p.stderr.read().decode()
            time.sleep(0.5)
        assert snaps, "no snapshot after SIGHUP"
        assert p.poll() is None, "SIGHUP must not stop training"
        p.send_signal(signal.SIGINT)
        _, err = p.communicate(timeout=300)
        assert p.returncode == 0, err.decode()


def test_sigint_effect_none_ignores_signal():
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_tiny_solver(tmp)
        p = start_train(solver, extra=("-sigint_effect=none",))
        time.sleep(3)
        p.send_signal(signal.SIGINT)
        time.sleep(2)
        assert p.poll() is None, "SIGINT with effect=none must be ignored"
        p.kill()
        p.communicate(timeout=300)
