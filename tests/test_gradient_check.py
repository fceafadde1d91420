"""Finite-difference gradient checks through the engine (CPU mode) — the
reference's own GradientChecker strategy
(include/caffe/test/test_gradient_check_util.hpp:19-70): objective
O = 0.5*sum(y^2) injected as top diff y, analytic grads from Backward,
numeric grads by central differences on the input blob (force_backward)
and on every learnable param.  Pins each layer's Backward_cpu against
nothing but its own Forward_cpu — independent of the oracle.
"""
import os
import sys
import tempfile

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))

import caffe_amd as ca  # noqa: E402

EPS = 1e-2
THRESH = 2e-3  # |a-n| <= THRESH * max(1, |a|, |n|)  (fp32 FD floor)


def build_net(body, shapes):
    """Net with Input layer tops in0,in1,... of the given shapes."""
    tops = "\n".join(f'  top: "in{i}"' for i in range(len(shapes)))
    shp = "\n".join(
        "    shape { " + " ".join(f"dim: {d}" for d in s) + " }"
        for s in shapes)
    text = f"""name: "gc"
force_backward: true
layer {{
  name: "input"
  type: "Input"
{tops}
  input_param {{
{shp}
  }}
}}
{body}
"""
    f = tempfile.NamedTemporaryFile("w", suffix=".prototxt", delete=False)
    f.write(text)
    f.close()
    net = ca.Net.from_file(f.name, phase=0)
    os.unlink(f.name)
    return net


def objective(net, out):
    net.forward()
    y = net.blob(out).astype(np.float64)
    return 0.5 * float((y * y).sum())


def check_gradients(net, out, perturb_inputs=("in0",), n_samples=12,
                    loss_top=False, eps=EPS, thresh=THRESH):
    rng = np.random.default_rng(7)
    net.forward()
    y = net.blob(out)
    if loss_top:
        # top IS the objective (scalar loss, weight 1)
        pass
    else:
        net.set_blob(out, y, diff=True)
    net.backward()

    def obj():
        if loss_top:
            net.forward()
            return float(net.blob(out).astype(np.float64).sum())
        return objective(net, out)

    # input blobs
    for name in perturb_inputs:
        analytic = net.blob(name, diff=True).ravel()
        x = net.blob(name).ravel().copy()
        idxs = rng.choice(x.size, size=min(n_samples, x.size),
                          replace=False)
        for i in idxs:
            xi = x[i]
            x[i] = xi + eps
            net.set_blob(name, x)
            op = obj()
            x[i] = xi - eps
            net.set_blob(name, x)
            om = obj()
            x[i] = xi
            net.set_blob(name, x)
            num = (op - om) / (2 * eps)
            a = analytic[i]
            assert abs(a - num) <= thresh * max(1.0, abs(a), abs(num)), \
                (name, i, a, num)
    # params (skip lr_mult-0 BN stats: indices with 'bn' stats are still
    # checked numerically-zero-safe since their diffs are zero)
    for pidx in range(net.num_params()):
        lname, bi, cnt = net.param_info(pidx)
        analytic = net.param(pidx, diff=True)
        w = net.param(pidx).copy()
        if analytic.size and not np.any(analytic):
            continue  # stats blobs / unused params
        idxs = rng.choice(cnt, size=min(n_samples, cnt), replace=False)
        for i in idxs:
            wi = w[i]
            w[i] = wi + eps
            net.set_param(pidx, w)
            op = obj()
            w[i] = wi - eps
            net.set_param(pidx, w)
            om = obj()
            w[i] = wi
            net.set_param(pidx, w)
            num = (op - om) / (2 * eps)
            a = analytic[i]
            assert abs(a - num) <= thresh * max(1.0, abs(a), abs(num)), \
                (lname, bi, i, a, num)


def seeded(shape, scale=1.0, seed=3):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal(shape) * scale).astype(np.float32)


def setup_function(_):
    ca.set_mode("cpu")
    ca.set_random_seed(17)


@pytest.mark.gpu
@pytest.mark.parametrize("case", ["conv", "bn", "pool", "ip"])
def test_grad_gpu(case):
    # same FD harness through the HIP kernels: pins each GPU backward
    # against its own GPU forward, independent of the CPU oracle
    ca.set_mode("gpu")
    ca.set_random_seed(17)
    {"conv": test_grad_conv,
     "bn": test_grad_batchnorm,
     "pool": test_grad_pool_max_and_ave,
     "ip": test_grad_inner_product}[case]()
    ca.set_mode("cpu")


def test_grad_conv():
    net = build_net("""layer { name: "c" type: "Convolution" bottom: "in0"
  top: "out" convolution_param { num_output: 4 kernel_size: 3 pad: 1
  weight_filler { type: "gaussian" std: 0.3 }
  bias_filler { type: "gaussian" std: 0.1 } } }""",
                    [(2, 3, 6, 6)])
    net.set_blob("in0", seeded((2, 3, 6, 6)))
    check_gradients(net, "out")


def test_grad_conv_stride_group_dilation():
    net = build_net("""layer { name: "c" type: "Convolution" bottom: "in0"
  top: "out" convolution_param { num_output: 4 kernel_size: 3 stride: 2
  pad: 2 group: 2 dilation: 2
  weight_filler { type: "gaussian" std: 0.3 } } }""",
                    [(2, 4, 9, 9)])
    net.set_blob("in0", seeded((2, 4, 9, 9)))
    check_gradients(net, "out")


def test_grad_inner_product():
    net = build_net("""layer { name: "ip" type: "InnerProduct"
  bottom: "in0" top: "out" inner_product_param { num_output: 5
  weight_filler { type: "gaussian" std: 0.3 }
  bias_filler { type: "gaussian" std: 0.1 } } }""",
                    [(3, 2, 4, 4)])
    net.set_blob("in0", seeded((3, 2, 4, 4)))
    check_gradients(net, "out")


def test_grad_batchnorm():
    net = build_net("""layer { name: "bn" type: "BatchNorm" bottom: "in0"
  top: "out" batch_norm_param { scale_bias: true } }""",
                    [(4, 3, 5, 5)])
    net.set_blob("in0", seeded((4, 3, 5, 5)))
    check_gradients(net, "out")


def test_grad_pool_max_and_ave():
    for pool in ("MAX", "AVE"):
        net = build_net(f"""layer {{ name: "p" type: "Pooling"
  bottom: "in0" top: "out" pooling_param {{ pool: {pool} kernel_size: 3
  stride: 2 }} }}""", [(2, 3, 7, 7)])
        # well-separated values keep the max selection stable under +-EPS
        rng = np.random.default_rng(9)
        x = rng.permutation(2 * 3 * 7 * 7).astype(np.float32)
        x = (x / x.size * 4 - 2).reshape(2, 3, 7, 7)
        net.set_blob("in0", x)
        check_gradients(net, "out")


def test_grad_lrn():
    net = build_net("""layer { name: "l" type: "LRN" bottom: "in0"
  top: "out" lrn_param { local_size: 3 alpha: 0.5 beta: 0.75 } }""",
                    [(2, 6, 4, 4)])
    net.set_blob("in0", seeded((2, 6, 4, 4)))
    check_gradients(net, "out")


def test_grad_eltwise_sum():
    net = build_net("""layer { name: "e" type: "Eltwise" bottom: "in0"
  bottom: "in1" top: "out" eltwise_param { operation: SUM coeff: 1.5
  coeff: -0.5 } }""", [(2, 3, 4, 4), (2, 3, 4, 4)])
    net.set_blob("in0", seeded((2, 3, 4, 4), seed=3))
    net.set_blob("in1", seeded((2, 3, 4, 4), seed=4))
    check_gradients(net, "out", perturb_inputs=("in0", "in1"))


def test_grad_relu():
    # direct bottom with kink exclusion, the reference's ReLU recipe
    # (GradientChecker kink=0, kink_range — skip elements near the kink)
    net = build_net("""layer { name: "r" type: "ReLU" bottom: "in0"
  top: "out" }""", [(2, 3, 5, 5)])
    x = seeded((2, 3, 5, 5))
    x[np.abs(x) < 5 * EPS] = 0.5  # keep every element off the kink
    net.set_blob("in0", x)
    check_gradients(net, "out")


def test_grad_fanout_split_accumulation():
    # one blob consumed by TWO layers: its diff must accumulate both
    # consumers' contributions (the reference's insert_splits semantics,
    # net.cpp InsertSplits — the GoogLeNet inception fan-out pattern)
    net = build_net("""layer { name: "a" type: "InnerProduct"
  bottom: "in0" top: "t1" inner_product_param { num_output: 4
  weight_filler { type: "gaussian" std: 0.3 } } }
layer { name: "b" type: "InnerProduct" bottom: "in0" top: "t2"
  inner_product_param { num_output: 4
  weight_filler { type: "gaussian" std: 0.3 } } }
layer { name: "e" type: "Eltwise" bottom: "t1" bottom: "t2" top: "out"
  eltwise_param { operation: SUM } }""", [(3, 6)])
    net.set_blob("in0", seeded((3, 6)))
    check_gradients(net, "out")


def test_grad_softmax_with_loss():
    net = build_net("""layer { name: "loss" type: "SoftmaxWithLoss"
  bottom: "in0" bottom: "in1" top: "out" }""",
                    [(4, 5), (4,)])
    net.set_blob("in0", seeded((4, 5)))
    net.set_blob("in1", np.array([0, 2, 4, 1], np.float32))
    check_gradients(net, "out", loss_top=True)
