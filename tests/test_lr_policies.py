"""Learning-rate policy parity (sgd_solver.cpp:24-66 + rampup :27-33):
run `caffe train` with display:1 for each lr_policy and compare the
printed per-iteration lr against the closed forms restated here.
"""
import math
import os
import re
import subprocess
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CAFFE = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")

NET = """net_param {
  name: "lrnet"
  layer {
    name: "input"
    type: "Input"
    top: "in0"
    top: "in1"
    input_param {
      shape { dim: 2 dim: 3 dim: 4 dim: 4 }
      shape { dim: 2 }
    }
  }
  layer {
    name: "ip"
    type: "InnerProduct"
    bottom: "in0"
    top: "fc"
    inner_product_param { num_output: 3 }
  }
  layer {
    name: "loss"
    type: "SoftmaxWithLoss"
    bottom: "fc"
    bottom: "in1"
    top: "loss"
  }
}
"""

ITERS = 8
BASE_LR = 0.5


def run_policy(extra):
    with tempfile.TemporaryDirectory() as tmp:
        solver = os.path.join(tmp, "s.prototxt")
        with open(solver, "w") as f:
            f.write(f"""base_lr: {BASE_LR}
max_iter: {ITERS}
display: 1
momentum: 0.0
snapshot_prefix: "{tmp}/x"
snapshot_after_train: false
{extra}
{NET}""")
        out = subprocess.run([CAFFE, "train", f"-solver={solver}"],
                             capture_output=True, text=True, timeout=300)
        assert out.returncode == 0, out.stderr
        lrs = [float(m.group(1)) for m in
               re.finditer(r"lr = ([0-9.eE+-]+)", out.stderr)]
        assert len(lrs) == ITERS, out.stderr
        return lrs  # lrs[i] is the lr used AT iteration i


def assert_close(lrs, expect):
    for i, (a, b) in enumerate(zip(lrs, expect)):
        assert abs(a - b) <= 1e-5 * max(1.0, abs(b)), (i, a, b, lrs)


def test_fixed():
    assert_close(run_policy('lr_policy: "fixed"'), [BASE_LR] * ITERS)


def test_step():
    lrs = run_policy('lr_policy: "step"\ngamma: 0.5\nstepsize: 3')
    assert_close(lrs, [BASE_LR * 0.5 ** (i // 3) for i in range(ITERS)])


def test_exp():
    lrs = run_policy('lr_policy: "exp"\ngamma: 0.9')
    assert_close(lrs, [BASE_LR * 0.9 ** i for i in range(ITERS)])


def test_inv():
    lrs = run_policy('lr_policy: "inv"\ngamma: 0.01\npower: 0.75')
    assert_close(lrs,
                 [BASE_LR * (1 + 0.01 * i) ** -0.75 for i in range(ITERS)])


def test_multistep():
    lrs = run_policy(
        'lr_policy: "multistep"\ngamma: 0.1\nstepvalue: 2\nstepvalue: 5')
    expect = [BASE_LR * 0.1 ** sum(i >= s for s in (2, 5))
              for i in range(ITERS)]
    assert_close(lrs, expect)


def test_poly():
    lrs = run_policy('lr_policy: "poly"\npower: 2.0')
    assert_close(lrs,
                 [BASE_LR * (1 - i / ITERS) ** 2.0 for i in range(ITERS)])


def test_sigmoid():
    lrs = run_policy('lr_policy: "sigmoid"\ngamma: -1.0\nstepsize: 4')
    assert_close(lrs,
                 [BASE_LR / (1 + math.exp(1.0 * (i - 4)))
                  for i in range(ITERS)])


def test_rampup():
    # rampup applies before any policy (sgd_solver.cpp:27-33): linear from
    # rampup_lr to base_lr over rampup_interval iterations
    lrs = run_policy(
        'lr_policy: "fixed"\nrampup_interval: 4\nrampup_lr: 0.1')
    expect = [0.1 + (BASE_LR - 0.1) * i / 4 if i < 4 else BASE_LR
              for i in range(ITERS)]
    assert_close(lrs, expect)
