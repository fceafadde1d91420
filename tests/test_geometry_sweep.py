"""Randomized conv/pool geometry parity: engine CPU vs oracle (two
independent implementations) over asymmetric kernels, strides, pads,
dilation-1 groups — combinations the fixed model-shape sets never hit.
Deterministic seeds; forward + input/weight gradients at the exact-f32
tolerance class.
"""
import numpy as np
import pytest

import caffe_amd as ca
from engine_util import net_from_text, relerr, run_layer
from oracle import oracle as orc

TOL = 1e-4


def rand_conv_cfg(rng):
    while True:
        C = int(rng.integers(1, 9))
        grp = int(rng.choice([1, 1, 1, 2, 4]))
        if C % grp:
            continue
        Co = grp * int(rng.integers(1, 5))
        kh = int(rng.integers(1, 6))
        kw = int(rng.integers(1, 6))
        sh = int(rng.integers(1, 4))
        sw = int(rng.integers(1, 4))
        ph = int(rng.integers(0, kh + 2))  # pad may exceed kernel (legal)
        pw = int(rng.integers(0, kw + 2))
        H = int(rng.integers(kh, kh + 12))
        W = int(rng.integers(kw, kw + 12))
        N = int(rng.integers(1, 4))
        return dict(N=N, C=C, H=H, W=W, Co=Co, kh=kh, kw=kw, sh=sh,
                    sw=sw, ph=ph, pw=pw, grp=grp)


@pytest.mark.parametrize("seed", range(16))
def test_conv_geometry_cpu(seed):
    rng = np.random.default_rng(3000 + seed)
    g = rand_conv_cfg(rng)
    ca.set_mode("cpu")
    x = rng.standard_normal((g["N"], g["C"], g["H"], g["W"])) \
        .astype(np.float32)
    w = (rng.standard_normal((g["Co"], g["C"] // g["grp"], g["kh"],
                              g["kw"])) * 0.3).astype(np.float32)
    body = f"""layer {{ name: "conv" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {g["Co"]}
  kernel_h: {g["kh"]} kernel_w: {g["kw"]} stride_h: {g["sh"]}
  stride_w: {g["sw"]} pad_h: {g["ph"]} pad_w: {g["pw"]}
  group: {g["grp"]} bias_term: false }} }}"""
    y_ref = orc.conv_fwd(x, w, None, pad=(g["ph"], g["pw"]),
                         stride=(g["sh"], g["sw"]), group=g["grp"])
    dy = rng.standard_normal(y_ref.shape).astype(np.float32)
    net, y = run_layer("cpu", [x.shape], body, [x], params=[w],
                       top_diff=dy)
    assert relerr(y, y_ref) < TOL, g
    dx_ref, dw_ref, _ = orc.conv_bwd(x, w, dy, pad=(g["ph"], g["pw"]),
                                     stride=(g["sh"], g["sw"]),
                                     group=g["grp"], want_db=False)
    assert relerr(net.blob("in0", diff=True), dx_ref) < TOL, g
    assert relerr(net.param(0, diff=True), dw_ref) < TOL, g


@pytest.mark.parametrize("seed", range(10))
def test_pool_geometry_cpu(seed):
    rng = np.random.default_rng(6000 + seed)
    kh = int(rng.integers(1, 5))
    kw = int(rng.integers(1, 5))
    sh = int(rng.integers(1, 4))
    sw = int(rng.integers(1, 4))
    ph = int(rng.integers(0, kh))  # reference: pad < kernel
    pw = int(rng.integers(0, kw))
    H = int(rng.integers(kh, kh + 10))
    W = int(rng.integers(kw, kw + 10))
    N, C = int(rng.integers(1, 3)), int(rng.integers(1, 5))
    mode = "MAX" if rng.integers(0, 2) else "AVE"
    ca.set_mode("cpu")
    x = rng.standard_normal((N, C, H, W)).astype(np.float32)
    body = f"""layer {{ name: "pool" type: "Pooling" bottom: "in0"
  top: "out" pooling_param {{ pool: {mode} kernel_h: {kh} kernel_w: {kw}
  stride_h: {sh} stride_w: {sw} pad_h: {ph} pad_w: {pw} }} }}"""
    if mode == "MAX":
        y_ref, _ = orc.pool_max_fwd(x, kh, kw, ph, pw, sh, sw)
    else:
        y_ref = orc.pool_ave_fwd(x, kh, kw, ph, pw, sh, sw)
    net, y = run_layer("cpu", [x.shape], body, [x])
    assert relerr(y, y_ref) < TOL, (kh, kw, sh, sw, ph, pw, H, W, mode)
