"""Scale layer (reference layers/scale_layer.cpp) — the BVLC-format
BatchNorm+Scale prototxt pattern: y = x*scale[c] (+bias[c]), channel
axis, in-place capable.  Checked vs torch, by finite differences
(incl. in-place), and as the full BVLC-style block.
"""
import numpy as np
import pytest
import torch

import caffe_amd as ca
from engine_util import relerr, run_layer, net_from_text
from test_gradient_check import build_net, check_gradients, seeded

TOL = 1e-4


def test_scale_vs_torch():
    shape = (3, 5, 4, 4)
    rng = np.random.default_rng(8)
    x = rng.standard_normal(shape).astype(np.float32)
    sc = (1 + 0.3 * rng.standard_normal(5)).astype(np.float32)
    bi = (0.2 * rng.standard_normal(5)).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)

    xt = torch.tensor(x, requires_grad=True)
    st = torch.tensor(sc, requires_grad=True)
    bt = torch.tensor(bi, requires_grad=True)
    yt = xt * st.view(1, -1, 1, 1) + bt.view(1, -1, 1, 1)
    yt.backward(torch.tensor(dy))

    body = """layer { name: "s" type: "Scale" bottom: "in0" top: "out"
  scale_param { bias_term: true } }"""
    net, out = run_layer("cpu", [shape], body, [x], params=[sc, bi],
                         top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < TOL
    assert relerr(net.param(0, diff=True), st.grad.numpy()) < TOL
    assert relerr(net.param(1, diff=True), bt.grad.numpy()) < TOL


def test_scale_default_is_identity():
    # no filler: scale fills with 1 (reference scale_layer.cpp default)
    shape = (2, 3, 4, 4)
    rng = np.random.default_rng(9)
    x = rng.standard_normal(shape).astype(np.float32)
    body = """layer { name: "s" type: "Scale" bottom: "in0"
  top: "out" }"""
    _, out = run_layer("cpu", [shape], body, [x])
    assert np.array_equal(out, x)


def test_scale_gradients_fd():
    net = build_net("""layer { name: "s" type: "Scale" bottom: "in0"
  top: "out" scale_param { bias_term: true
  filler { type: "gaussian" std: 0.5 }
  bias_filler { type: "gaussian" std: 0.1 } } }""", [(2, 4, 5, 5)])
    net.set_blob("in0", seeded((2, 4, 5, 5)))
    check_gradients(net, "out")


def test_bvlc_style_bn_scale_block_trains():
    # the BVLC-format pattern: stats-only BatchNorm + in-place Scale
    ca.set_mode("cpu")
    ca.set_random_seed(31)
    solver = ca.Solver(text="""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 31
net_param {
  name: "bvlc_block"
  layer {
    name: "input"
    type: "Input"
    top: "in0"
    top: "in1"
    input_param {
      shape { dim: 4 dim: 3 dim: 8 dim: 8 }
      shape { dim: 4 }
    }
  }
  layer {
    name: "c1"
    type: "Convolution"
    bottom: "in0"
    top: "c1"
    convolution_param { num_output: 4 kernel_size: 3 pad: 1
      weight_filler { type: "msra" } }
  }
  layer { name: "bn1" type: "BatchNorm" bottom: "c1" top: "bn1" }
  layer {
    name: "scale1"
    type: "Scale"
    bottom: "bn1"
    top: "bn1"
    scale_param { bias_term: true }
  }
  layer { name: "r1" type: "ReLU" bottom: "bn1" top: "bn1" }
  layer {
    name: "ip"
    type: "InnerProduct"
    bottom: "bn1"
    top: "fc"
    inner_product_param { num_output: 5
      weight_filler { type: "xavier" } }
  }
  layer {
    name: "loss"
    type: "SoftmaxWithLoss"
    bottom: "fc"
    bottom: "in1"
    top: "loss"
  }
}
""")
    rng = np.random.default_rng(3)
    losses = []
    for it in range(6):
        solver.net.set_blob("in0", rng.standard_normal(
            (4, 3, 8, 8)).astype(np.float32))
        solver.net.set_blob("in1",
                            rng.integers(0, 5, 4).astype(np.float32))
        solver.step(1)
        losses.append(solver.loss())
    assert all(np.isfinite(l) for l in losses)
    # scale param moved (it is learnable)
    p = {i: solver.net.param_info(i)
         for i in range(solver.net.num_params())}
    assert any(n == "scale1" for n, _, _ in p.values()), p


def test_bias_vs_torch():
    shape = (2, 4, 5, 5)
    rng = np.random.default_rng(21)
    x = rng.standard_normal(shape).astype(np.float32)
    bi = (0.3 * rng.standard_normal(4)).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)
    xt = torch.tensor(x, requires_grad=True)
    bt = torch.tensor(bi, requires_grad=True)
    yt = xt + bt.view(1, -1, 1, 1)
    yt.backward(torch.tensor(dy))
    body = """layer { name: "b" type: "Bias" bottom: "in0"
  top: "out" }"""
    net, out = run_layer("cpu", [shape], body, [x], params=[bi],
                         top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < TOL
    assert relerr(net.param(0, diff=True), bt.grad.numpy()) < TOL


@pytest.mark.gpu
def test_bias_gpu_vs_cpu():
    shape = (2, 5, 4, 4)
    rng = np.random.default_rng(22)
    x = rng.standard_normal(shape).astype(np.float32)
    bi = (0.3 * rng.standard_normal(5)).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)
    body = """layer { name: "b" type: "Bias" bottom: "in0"
  top: "out" }"""
    res = {}
    for mode in ("cpu", "gpu"):
        net, out = run_layer(mode, [shape], body, [x], params=[bi],
                             top_diff=dy)
        res[mode] = (out, net.blob("in0", diff=True),
                     net.param(0, diff=True))
    for a, b in zip(res["cpu"], res["gpu"]):
        assert relerr(b, a) < TOL


@pytest.mark.gpu
def test_scale_gpu_vs_cpu():
    shape = (3, 6, 5, 5)
    rng = np.random.default_rng(12)
    x = rng.standard_normal(shape).astype(np.float32)
    sc = (1 + 0.3 * rng.standard_normal(6)).astype(np.float32)
    bi = (0.2 * rng.standard_normal(6)).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)
    body = """layer { name: "s" type: "Scale" bottom: "in0" top: "out"
  scale_param { bias_term: true } }"""
    res = {}
    for mode in ("cpu", "gpu"):
        net, out = run_layer(mode, [shape], body, [x], params=[sc, bi],
                             top_diff=dy)
        res[mode] = (out, net.blob("in0", diff=True),
                     net.param(0, diff=True), net.param(1, diff=True))
    for a, b in zip(res["cpu"], res["gpu"]):
        assert relerr(b, a) < TOL
