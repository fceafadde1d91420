"""CLI (`caffe train/test/time`) + snapshot/restore round trip, CPU mode.

Covers the reference surface SURVEY.md §8b names: the caffe CLI
(tools/caffe.cpp:28-46) and the .caffemodel/.solverstate binaryproto
snapshot (solver.cpp:542-604).
"""
import os
import subprocess
import sys
import tempfile

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CAFFE = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))


def make_lenet_solver(tmp, extra=""):
    os.makedirs(os.path.join(REPO, "models", "generated"), exist_ok=True)
    subprocess.check_call([sys.executable,
                           os.path.join(REPO, "models", "gen_models.py")])
    net = os.path.join(REPO, "models", "generated",
                       "lenet_train_val.prototxt")
    solver = os.path.join(tmp, "solver.prototxt")
    with open(solver, "w") as f:
        f.write(f'''net: "{net}"
base_lr: 0.01
lr_policy: "fixed"
momentum: 0.9
weight_decay: 0.0005
display: 0
max_iter: 4
snapshot_prefix: "{tmp}/lenet"
random_seed: 7
test_interval: 2
test_iter: 2
{extra}
''')
    return solver


def run_env():
    env = dict(os.environ)
    env["CAFFE_AMD_SYN"] = "1"
    return env


def test_cli_train_test_time_and_restore():
    import caffe_amd as ca
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_lenet_solver(tmp)
        # train via CLI (CPU, synthetic MNIST shape set through the C API is
        # not available to the binary — LeNet has no crop_size, so drive
        # through the python mirror instead for shape control)
        ca.set_mode("cpu")
        ca.set_synthetic_shape(1, 28, 28, 10)
        ca.set_random_seed(7)
        s = ca.Solver(path=solver)
        s.step(4)
        _ck(ca._lib.caffe_solver_snapshot(s._h))
        model = os.path.join(tmp, "lenet_iter_4.caffemodel")
        state = os.path.join(tmp, "lenet_iter_4.solverstate")
        assert os.path.exists(model) and os.path.exists(state)

        # weights round trip: fresh solver restored from state must produce
        # identical params and continue from iter 4
        s2 = ca.Solver(path=solver)
        _ck(ca._lib.caffe_solver_restore(s2._h, state.encode()))
        assert s2.iter == 4
        for i in range(s.net.num_params()):
            a = s.net.param(i)
            b = s2.net.param(i)
            assert np.array_equal(a, b), f"param {i} differs after restore"
        # momentum history must also match: one more identical-data step
        s.step(1)
        s2.step(1)
        for i in range(s.net.num_params()):
            assert np.allclose(s.net.param(i), s2.net.param(i),
                               rtol=1e-6, atol=1e-7)


def _ck(rc):
    assert rc == 0


def test_cli_binary_train_then_test():
    """The full `caffe train` → snapshot → `caffe test -weights` chain
    through the BINARY (tools/caffe.cpp:213/:291 surfaces): train writes
    lenet_iter_4.caffemodel, test loads it on the TEST phase net and
    prints averaged scores (accuracy + loss tops)."""
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_lenet_solver(tmp)
        net = os.path.join(REPO, "models", "generated",
                           "lenet_train_val.prototxt")
        env = dict(os.environ, CAFFE_SYN_SHAPE="1x28x28x10")
        out = subprocess.run(
            [CAFFE, "train", f"-solver={solver}"],
            capture_output=True, text=True, timeout=600, env=env)
        assert out.returncode == 0, out.stderr
        assert "Optimization Done." in out.stderr
        model = os.path.join(tmp, "lenet_iter_4.caffemodel")
        assert os.path.exists(model), out.stderr
        out = subprocess.run(
            [CAFFE, "test", f"-model={net}", f"-weights={model}",
             "-iterations=2"],
            capture_output=True, text=True, timeout=600, env=env)
        assert out.returncode == 0, out.stderr
        scores = dict(line.split(" = ") for line in
                      out.stdout.strip().splitlines() if " = " in line)
        assert "accuracy" in scores and "loss" in scores, out.stdout
        assert 0.0 <= float(scores["accuracy"]) <= 1.0
        assert float(scores["loss"]) > 0.0


def test_cli_binary_time_and_device_query():
    # `caffe time` on the LeNet model, CPU
    with tempfile.TemporaryDirectory() as tmp:
        make_lenet_solver(tmp)
        net = os.path.join(REPO, "models", "generated",
                           "lenet_train_val.prototxt")
        out = subprocess.run(
            [CAFFE, "time", f"-model={net}", "-iterations=1"],
            capture_output=True, text=True, timeout=600,
            env=dict(os.environ, CAFFE_SYN_SHAPE="1x28x28x10"))
        assert out.returncode == 0, out.stderr
        assert "TOTAL" in out.stderr
        # per-layer forward/backward lines for every LeNet layer
        for lname in ("conv1", "pool1", "ip1", "loss"):
            assert lname in out.stderr, (lname, out.stderr)


def test_cli_finetune_weights_resets_iter():
    # `caffe train -weights=model` (finetune flow, tools/caffe.cpp:176):
    # params load from the .caffemodel but the iteration counter starts
    # at 0 (unlike -snapshot, which restores iter + momentum history)
    with tempfile.TemporaryDirectory() as tmp:
        solver = make_lenet_solver(tmp)
        env = dict(os.environ, CAFFE_SYN_SHAPE="1x28x28x10")
        out = subprocess.run([CAFFE, "train", f"-solver={solver}"],
                             capture_output=True, text=True, timeout=600,
                             env=env)
        assert out.returncode == 0, out.stderr
        model = os.path.join(tmp, "lenet_iter_4.caffemodel")
        assert os.path.exists(model)
        out = subprocess.run(
            [CAFFE, "train", f"-solver={solver}", f"-weights={model}",
             "-iterations=2"],
            capture_output=True, text=True, timeout=600, env=env)
        assert out.returncode == 0, out.stderr
        # finetune run counts from 0: snapshot_after_train writes iter_2
        assert os.path.exists(os.path.join(tmp,
                                           "lenet_iter_2.caffemodel")), \
            out.stderr


def test_cli_usage_and_unknown_command():
    out = subprocess.run([CAFFE], capture_output=True, text=True,
                         timeout=60)
    assert out.returncode == 1
    assert "usage:" in out.stderr
    out = subprocess.run([CAFFE, "frobnicate"], capture_output=True,
                         text=True, timeout=60)
    assert out.returncode == 1
    assert "unknown command" in out.stderr


def test_cli_train_missing_solver_flag():
    out = subprocess.run([CAFFE, "train"], capture_output=True, text=True,
                         timeout=60)
    assert out.returncode == 1
    assert "solver" in out.stderr


def test_cli_bad_invocations():
    # every bad invocation must exit 1 with a message — never crash
    # (signal) or hang
    import subprocess
    cases = [
        ([], b"usage"),
        (["bogus"], b"unknown command"),
        (["train"], b""),                       # no -solver
        (["train", "-solver=/does/not/exist.prototxt"], b""),
        (["test", "-model=/does/not/exist.prototxt"], b""),
        (["time"], b""),                        # no -model
    ]
    for args, expect in cases:
        r = subprocess.run([CAFFE] + args, capture_output=True, timeout=60)
        assert r.returncode == 1, (args, r.returncode,
                                   r.stderr.decode()[-200:])
        if expect:
            assert expect in r.stderr.lower(), (args, r.stderr[-200:])
