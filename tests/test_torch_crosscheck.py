"""Randomized cross-check against torch-CPU — the secondary oracle
SURVEY.md §8c names (never the implementation).  Seeded random layer
configs; forward AND backward (input + param grads via torch autograd)
compared at the reference's 1e-4 relative fp32 bar.  Catches any
systematic error the hand-written oracle and the engine might share.
"""
import numpy as np
import pytest
import torch

from engine_util import relerr, run_layer

torch.manual_seed(0)

TOL = 1e-4
N_CASES = 8


def conv_cases():
    rng = np.random.default_rng(42)
    cases = []
    for _ in range(N_CASES):
        group = int(rng.choice([1, 1, 2]))
        cin = int(rng.integers(1, 5)) * group
        cout = int(rng.integers(1, 5)) * group
        k = int(rng.choice([1, 2, 3, 5]))
        s = int(rng.choice([1, 1, 2]))
        d = int(rng.choice([1, 1, 2])) if s == 1 else 1
        p = int(rng.integers(0, k))
        h = int(rng.integers(k * d + 1, 14))
        n = int(rng.integers(1, 4))
        cases.append((n, cin, h, h, cout, k, s, p, d, group))
    return cases


@pytest.mark.parametrize("case", conv_cases(),
                         ids=lambda c: "x".join(map(str, c)))
def test_conv_vs_torch(case):
    n, cin, h, w, cout, k, s, p, d, group = case
    rng = np.random.default_rng(hash(case) % 2**32)
    x = rng.standard_normal((n, cin, h, w)).astype(np.float32)
    wgt = rng.standard_normal(
        (cout, cin // group, k, k)).astype(np.float32) * 0.5
    b = rng.standard_normal(cout).astype(np.float32) * 0.1

    body = f"""layer {{ name: "c" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {cout} kernel_size: {k}
  stride: {s} pad: {p} dilation: {d} group: {group} }} }}"""
    # torch reference with autograd
    xt = torch.tensor(x, requires_grad=True)
    wt = torch.tensor(wgt, requires_grad=True)
    bt = torch.tensor(b, requires_grad=True)
    yt = torch.nn.functional.conv2d(xt, wt, bt, stride=s, padding=p,
                                    dilation=d, groups=group)
    dy = np.asarray(
        np.random.default_rng(7).standard_normal(tuple(yt.shape)),
        np.float32)
    yt.backward(torch.tensor(dy))

    net, out = run_layer("cpu", [(n, cin, h, w)], body, [x],
                         params=[wgt, b], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < TOL
    assert relerr(net.param(0, diff=True), wt.grad.numpy().ravel()) < TOL
    assert relerr(net.param(1, diff=True), bt.grad.numpy().ravel()) < TOL


@pytest.mark.parametrize("kh,kw,ph,pw", [(1, 7, 0, 3), (7, 1, 3, 0),
                                         (3, 5, 1, 2)])
def test_conv_asymmetric_vs_torch(kh, kw, ph, pw):
    # kernel_h/kernel_w + pad_h/pad_w — the inception_v3 1x7/7x1 pattern
    n, cin, h, w, cout = 2, 3, 9, 9, 4
    rng = np.random.default_rng(kh * 10 + kw)
    x = rng.standard_normal((n, cin, h, w)).astype(np.float32)
    wgt = rng.standard_normal((cout, cin, kh, kw)).astype(np.float32) * 0.4
    b = rng.standard_normal(cout).astype(np.float32) * 0.1

    xt = torch.tensor(x, requires_grad=True)
    wt = torch.tensor(wgt, requires_grad=True)
    bt = torch.tensor(b, requires_grad=True)
    yt = torch.nn.functional.conv2d(xt, wt, bt, padding=(ph, pw))
    dy = rng.standard_normal(tuple(yt.shape)).astype(np.float32)
    yt.backward(torch.tensor(dy))

    body = f"""layer {{ name: "c" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {cout} kernel_h: {kh}
  kernel_w: {kw} pad_h: {ph} pad_w: {pw} }} }}"""
    net, out = run_layer("cpu", [(n, cin, h, w)], body, [x],
                         params=[wgt, b], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < TOL
    assert relerr(net.param(0, diff=True), wt.grad.numpy().ravel()) < TOL
    assert relerr(net.param(1, diff=True), bt.grad.numpy().ravel()) < TOL


@pytest.mark.gpu
def test_conv_asymmetric_gpu_vs_torch():
    # the HIP explicit-col path with kernel_h != kernel_w, vs torch-CPU
    n, cin, h, w, cout, kh, kw, ph, pw = 2, 3, 9, 9, 4, 1, 7, 0, 3
    rng = np.random.default_rng(3)
    x = rng.standard_normal((n, cin, h, w)).astype(np.float32)
    wgt = rng.standard_normal((cout, cin, kh, kw)).astype(np.float32) * 0.4
    b = rng.standard_normal(cout).astype(np.float32) * 0.1
    xt = torch.tensor(x, requires_grad=True)
    wt = torch.tensor(wgt, requires_grad=True)
    bt = torch.tensor(b, requires_grad=True)
    yt = torch.nn.functional.conv2d(xt, wt, bt, padding=(ph, pw))
    dy = rng.standard_normal(tuple(yt.shape)).astype(np.float32)
    yt.backward(torch.tensor(dy))
    body = f"""layer {{ name: "c" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {cout} kernel_h: {kh}
  kernel_w: {kw} pad_h: {ph} pad_w: {pw} }} }}"""
    net, out = run_layer("gpu", [(n, cin, h, w)], body, [x],
                         params=[wgt, b], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.param(0, diff=True), wt.grad.numpy().ravel()) < TOL


@pytest.mark.parametrize("shape", [(2, 3, 6, 6), (4, 8, 5, 5), (1, 16, 3, 3)])
def test_batchnorm_vs_torch(shape):
    n, c, h, w = shape
    rng = np.random.default_rng(c)
    x = rng.standard_normal(shape).astype(np.float32)
    sc = (1 + 0.1 * rng.standard_normal(c)).astype(np.float32)
    bi = (0.1 * rng.standard_normal(c)).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)

    xt = torch.tensor(x, requires_grad=True)
    st = torch.tensor(sc, requires_grad=True)
    bt = torch.tensor(bi, requires_grad=True)
    yt = torch.nn.functional.batch_norm(
        xt, None, None, st, bt, training=True, eps=1e-5)
    yt.backward(torch.tensor(dy))

    body = """layer { name: "bn" type: "BatchNorm" bottom: "in0"
  top: "out" batch_norm_param { scale_bias: true eps: 1e-5 } }"""
    # the param API exposes learnables only: 0 = scale (blob 3),
    # 1 = bias (blob 4); the stats blobs are lr_mult-0 internals
    net, out = run_layer("cpu", [shape], body, [x],
                         params=[sc, bi], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < 5 * TOL
    assert relerr(net.param(0, diff=True), st.grad.numpy()) < TOL
    assert relerr(net.param(1, diff=True), bt.grad.numpy()) < TOL


@pytest.mark.parametrize("size,alpha,beta", [(5, 1e-4, 0.75), (3, 0.5, 0.5)])
def test_lrn_vs_torch(size, alpha, beta):
    # caffe cross-channel LRN == torch local_response_norm (both use
    # 1 + alpha/size * sum within the clipped channel window, k=1)
    shape = (2, 8, 6, 6)
    rng = np.random.default_rng(size)
    x = rng.standard_normal(shape).astype(np.float32)
    dy = rng.standard_normal(shape).astype(np.float32)
    xt = torch.tensor(x, requires_grad=True)
    yt = torch.nn.functional.local_response_norm(xt, size, alpha=alpha,
                                                 beta=beta, k=1.0)
    yt.backward(torch.tensor(dy))
    body = f"""layer {{ name: "l" type: "LRN" bottom: "in0" top: "out"
  lrn_param {{ local_size: {size} alpha: {alpha} beta: {beta} }} }}"""
    net, out = run_layer("cpu", [shape], body, [x], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < 5 * TOL


@pytest.mark.parametrize("pool,shape,k,s", [
    ("MAX", (2, 3, 9, 9), 3, 2), ("AVE", (2, 3, 8, 8), 2, 2),
    ("MAX", (1, 4, 7, 7), 2, 1)])
def test_pool_vs_torch(pool, shape, k, s):
    rng = np.random.default_rng(k * 10 + s)
    # distinct values keep torch's and the engine's max tie-breaks aligned
    x = rng.permutation(np.prod(shape)).astype(np.float32).reshape(shape)
    xt = torch.tensor(x, requires_grad=True)
    if pool == "MAX":
        yt = torch.nn.functional.max_pool2d(xt, k, stride=s,
                                            ceil_mode=True)
    else:
        yt = torch.nn.functional.avg_pool2d(xt, k, stride=s,
                                            ceil_mode=True)
    dy = rng.standard_normal(tuple(yt.shape)).astype(np.float32)
    yt.backward(torch.tensor(dy))

    body = f"""layer {{ name: "p" type: "Pooling" bottom: "in0"
  top: "out" pooling_param {{ pool: {pool} kernel_size: {k}
  stride: {s} }} }}"""
    net, out = run_layer("cpu", [shape], body, [x], top_diff=dy)
    assert relerr(out, yt.detach().numpy()) < TOL
    assert relerr(net.blob("in0", diff=True), xt.grad.numpy()) < TOL
