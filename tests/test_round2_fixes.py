"""Round-2 correctness regressions:

- SolverState history blobs travel in FORWARD learnable-param order (the
  reference's order, sgd_solver.cpp:262-353) even though the engine's
  arena is reverse-layer order; a reference-generated .solverstate must
  map onto the right params.
- Accuracy reports 0 (not 1.0) on NaN predictions: the strictly-greater
  rank test is vacuously 0 for NaN (reference accuracy_layer.cpp shares
  the form, but an engine must not present NaN as a perfect score).
- In-place BatchNorm (top==bottom, the BVLC prototxt idiom) trains with
  correct gradients: backward recomputes x-hat from a saved copy of the
  input, which the in-place forward would otherwise have overwritten.
"""
import os
import tempfile

import numpy as np
import pytest

import caffe_amd as ca
from engine_util import TOL, input_net, net_from_text, relerr

from test_caffemodel_wire import decode_blob, walk


def test_solverstate_history_forward_order():
    # conv (W 4x3x3x3 = 108, b 4) then ip (W 5xK, b 5): forward order on
    # the wire means hist[0] is the CONV weight — the arena order would
    # put the IP bias first
    ca.set_mode("cpu")
    tmp = tempfile.mkdtemp()
    solver = ca.Solver(text="""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 7
snapshot_prefix: "PFX/s"
net_param {
  name: "s"
  layer { name: "input" type: "Input" top: "in0" top: "in1"
    input_param { shape { dim: 2 dim: 3 dim: 6 dim: 6 } shape { dim: 2 } } }
  layer { name: "c1" type: "Convolution" bottom: "in0" top: "mid"
    convolution_param { num_output: 4 kernel_size: 3
      weight_filler { type: "gaussian" std: 0.2 }
      bias_filler { type: "constant" value: 0.1 } } }
  layer { name: "ip" type: "InnerProduct" bottom: "mid" top: "fc"
    inner_product_param { num_output: 5
      weight_filler { type: "gaussian" std: 0.2 }
      bias_filler { type: "constant" } } }
  layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "in1"
    top: "loss" }
}
""".replace("PFX", tmp))
    rng = np.random.default_rng(3)
    solver.net.set_blob("in0",
                        rng.standard_normal((2, 3, 6, 6)).astype(np.float32))
    solver.net.set_blob("in1", np.array([0, 3], np.float32))
    solver.step(2)
    assert ca._lib.caffe_solver_snapshot(solver._h) == 0
    raw = open(os.path.join(tmp, "s_iter_2.solverstate"), "rb").read()
    hist = [decode_blob(v) for f, w, v in walk(raw) if f == 3 and w == 2]
    shapes = [tuple(int(d) for d in s) for s, _ in hist]
    assert shapes == [(4, 3, 3, 3), (4,), (5, 4 * 4 * 4), (5,)], shapes
    # the conv-weight history must be the actual conv momentum: nonzero and
    # distinct from the ip history (sizes already distinguish here, but pin
    # values too — restore into a fresh solver and compare a further step)
    assert float(np.abs(hist[0][1]).max()) > 0

    # restore into a fresh solver (same permutation applied on read) and
    # verify resumed training matches uninterrupted training exactly
    state = os.path.join(tmp, "s_iter_2.solverstate")
    s2 = ca.Solver(text=open_solver_text(tmp))
    assert ca._lib.caffe_solver_restore(s2._h, state.encode()) == 0
    assert s2.iter == 2
    x = rng.standard_normal((2, 3, 6, 6)).astype(np.float32)
    for s in (solver, s2):
        s.net.set_blob("in0", x)
        s.net.set_blob("in1", np.array([0, 3], np.float32))
        s.step(1)
    for i in range(solver.net.num_params()):
        assert relerr(s2.net.param(i), solver.net.param(i)) < 1e-6


def open_solver_text(tmp):
    return """base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 7
snapshot_prefix: "PFX/s"
net_param {
  name: "s"
  layer { name: "input" type: "Input" top: "in0" top: "in1"
    input_param { shape { dim: 2 dim: 3 dim: 6 dim: 6 } shape { dim: 2 } } }
  layer { name: "c1" type: "Convolution" bottom: "in0" top: "mid"
    convolution_param { num_output: 4 kernel_size: 3
      weight_filler { type: "gaussian" std: 0.2 }
      bias_filler { type: "constant" value: 0.1 } } }
  layer { name: "ip" type: "InnerProduct" bottom: "mid" top: "fc"
    inner_product_param { num_output: 5
      weight_filler { type: "gaussian" std: 0.2 }
      bias_filler { type: "constant" } } }
  layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "in1"
    top: "loss" }
}
""".replace("PFX", tmp)


def test_accuracy_nan_is_incorrect():
    ca.set_mode("cpu")
    body = """layer { name: "acc" type: "Accuracy" bottom: "in0"
  bottom: "in1" top: "out" }"""
    net = net_from_text(input_net([(4, 10), (4,)], body))
    pred = np.full((4, 10), np.nan, np.float32)
    net.set_blob("in0", pred)
    net.set_blob("in1", np.array([1, 2, 3, 4], np.float32))
    net.forward()
    assert net.blob("out")[0] == 0.0  # NaN rows are never correct

    # half the rows NaN: only the finite ones can score
    pred = np.zeros((4, 10), np.float32)
    pred[0, 1] = 5.0   # correct
    pred[1, 0] = 5.0   # wrong (label 2)
    pred[2:] = np.nan
    net.set_blob("in0", pred)
    net.forward()
    assert abs(net.blob("out")[0] - 0.25) < 1e-6


def _bn_pair_net(inplace):
    mid_top = "mid" if inplace else "bnout"
    return f"""name: "t"
force_backward: true
layer {{ name: "input" type: "Input" top: "in0"
  input_param {{ shape {{ dim: 3 dim: 4 dim: 5 dim: 5 }} }} }}
layer {{ name: "c1" type: "Convolution" bottom: "in0" top: "mid"
  convolution_param {{ num_output: 4 kernel_size: 3 pad: 1
    weight_filler {{ type: "gaussian" std: 0.3 }} }} }}
layer {{ name: "bn" type: "BatchNorm" bottom: "mid" top: "{mid_top}"
  batch_norm_param {{ eps: 0.001 scale_bias: true }} }}
layer {{ name: "ip" type: "InnerProduct" bottom: "{mid_top}" top: "out"
  inner_product_param {{ num_output: 3
    weight_filler {{ type: "gaussian" std: 0.2 }} }} }}
"""


@pytest.mark.parametrize("mode", ["cpu"])
def test_inplace_bn_gradients(mode):
    # the in-place net's gradients must equal the out-of-place net's —
    # backward needs the ORIGINAL conv output, which in-place BN overwrites
    ca.set_mode(mode)
    rng = np.random.default_rng(11)
    x = rng.standard_normal((3, 4, 5, 5)).astype(np.float32)
    dy = rng.standard_normal((3, 3)).astype(np.float32)
    grads = {}
    for inplace in (False, True):
        ca.set_random_seed(5)
        net = net_from_text(_bn_pair_net(inplace))
        net.set_blob("in0", x)
        net.forward()
        net.set_blob("out", dy, diff=True)
        net.backward()
        grads[inplace] = [net.param(i, diff=True)
                          for i in range(net.num_params())]
        if not inplace:
            outs = net.blob("out").copy()
        else:
            assert relerr(net.blob("out"), outs) < TOL
    for a, b in zip(grads[False], grads[True]):
        assert relerr(b, a) < TOL


@pytest.mark.gpu
def test_inplace_bn_gradients_gpu():
    # GPU in-place BN vs CPU out-of-place: parity on every param grad
    rng = np.random.default_rng(11)
    x = rng.standard_normal((3, 4, 5, 5)).astype(np.float32)
    dy = rng.standard_normal((3, 3)).astype(np.float32)
    grads = {}
    for mode in ("cpu", "gpu"):
        ca.set_mode(mode)
        ca.set_random_seed(5)
        net = net_from_text(_bn_pair_net(True))
        net.set_blob("in0", x)
        net.forward()
        net.set_blob("out", dy, diff=True)
        net.backward()
        grads[mode] = [net.param(i, diff=True)
                       for i in range(net.num_params())]
    for a, b in zip(grads["cpu"], grads["gpu"]):
        # absolute floor: the conv BIAS grad is a numerical zero here (BN
        # backward's dx sums to ~0 per channel), so pure relative error
        # compares rounding noise
        denom = max(float(np.abs(a).max()), 1.0)
        assert float(np.abs(np.asarray(a) - b).max()) < TOL * denom


def _loss_net():
    return input_net(
        [(4, 7), (4,)],
        """layer { name: "loss" type: "SoftmaxWithLoss" bottom: "in0"
  bottom: "in1" top: "out" }""")


@pytest.mark.gpu
@pytest.mark.parametrize("poison", ["nan", "pinf", "ninf"])
def test_nonfinite_softmaxloss_gpu(poison):
    # CPU and GPU must agree on non-finite logits: NaN / +inf rows yield
    # NaN loss on BOTH paths; -inf logits are benign (prob 0)
    rng = np.random.default_rng(4)
    x = rng.standard_normal((4, 7)).astype(np.float32)
    val = dict(nan=np.nan, pinf=np.inf, ninf=-np.inf)[poison]
    x[1, 3] = val
    labels = np.array([0, 1, 2, 3], np.float32)
    out = {}
    for mode in ("cpu", "gpu"):
        ca.set_mode(mode)
        net = net_from_text(_loss_net())
        net.set_blob("in0", x)
        net.set_blob("in1", labels)
        net.forward()
        out[mode] = float(net.blob("out")[0])
    if poison in ("nan", "pinf"):
        assert np.isnan(out["cpu"]), out
        assert np.isnan(out["gpu"]), out
    else:
        assert np.isfinite(out["cpu"]) and np.isfinite(out["gpu"]), out
        assert abs(out["cpu"] - out["gpu"]) < 1e-5 * max(1, abs(out["cpu"]))


@pytest.mark.gpu
def test_nan_softmax_row_isolation_gpu():
    # a poisoned row must not leak into other rows, and the poisoned row
    # must be NaN on both paths (Softmax layer, not the loss)
    rng = np.random.default_rng(9)
    x = rng.standard_normal((3, 5)).astype(np.float32)
    x[1, 2] = np.nan
    body = """layer { name: "sm" type: "Softmax" bottom: "in0"
  top: "out" }"""
    out = {}
    for mode in ("cpu", "gpu"):
        ca.set_mode(mode)
        net = net_from_text(input_net([(3, 5)], body))
        net.set_blob("in0", x)
        net.forward()
        out[mode] = net.blob("out").reshape(3, 5)
    for mode in ("cpu", "gpu"):
        assert np.isnan(out[mode][1]).all(), (mode, out[mode])
        assert np.isfinite(out[mode][[0, 2]]).all(), (mode, out[mode])
    assert relerr(out["gpu"][[0, 2]], out["cpu"][[0, 2]]) < TOL
