"""bf16 mixed-precision GEMM parity (GPU): conv/IP contractions computed
by the bf16 MFMA kernel (fp32 storage, fp32 accumulation) against the
fp32 oracle at the bf16 tolerance class — bf16 has an 8-bit mantissa, so
inputs round at ~0.4% relative; with fp32 accumulation the per-blob
relative error stays ~1e-2 (the reference's fp16 runs used the same
reduced-precision tolerance class, e.g. alexnet_pfp16.log).  The engine's
default stays exact f32 — these tests flip the mode per-case and restore.
"""
import numpy as np
import pytest

import caffe_amd as ca
from engine_util import input_net, net_from_text, run_layer
from layer_checks import rng
from oracle import oracle as orc

pytestmark = pytest.mark.gpu

BTOL = 3e-2  # bf16-input GEMM tolerance (fp32 accumulation)


@pytest.fixture(autouse=True)
def bf16_mode():
    ca.set_mode("gpu")
    ca.set_compute("bf16")
    yield
    ca.set_compute("f32")


def _relerr(a, b):
    a = np.asarray(a, np.float64).ravel()
    b = np.asarray(b, np.float64).ravel()
    assert a.size == b.size, (a.size, b.size)
    return np.abs(a - b).max() / max(np.abs(b).max(), 1e-6)


@pytest.mark.parametrize("cfg", [
    dict(N=2, C=8, H=13, W=13, Co=16, k=3, s=1, p=1),          # explicit
    dict(N=2, C=4, H=32, W=32, Co=8, k=3, s=1, p=1),           # implicit
    dict(N=2, C=8, H=16, W=16, Co=72, k=1, s=1, p=0),          # 1x1 view
    dict(N=2, C=3, H=55, W=55, Co=8, k=7, s=2, p=3),           # strided view
    dict(N=3, C=5, H=11, W=17, Co=7, k=5, s=3, p=2),           # odd explicit
])
def test_conv_bf16(cfg):
    N, C, H, W = cfg["N"], cfg["C"], cfg["H"], cfg["W"]
    Co, k, s, p = cfg["Co"], cfg["k"], cfg["s"], cfg["p"]
    x = rng.standard_normal((N, C, H, W)).astype(np.float32)
    w = (rng.standard_normal((Co, C, k, k)) * 0.2).astype(np.float32)
    body = f"""layer {{ name: "conv" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {Co} kernel_size: {k}
  stride: {s} pad: {p} bias_term: false }} }}"""
    y_ref = orc.conv_fwd(x, w, None, pad=(p, p), stride=(s, s))
    dy = rng.standard_normal(y_ref.shape).astype(np.float32)
    net, y = run_layer("gpu", [(N, C, H, W)], body, [x], params=[w],
                       top_diff=dy)
    assert _relerr(y, y_ref) < BTOL, f"fwd {_relerr(y, y_ref)}"
    dx_ref, dw_ref, _ = orc.conv_bwd(x, w, dy, pad=(p, p), stride=(s, s),
                                     want_db=False)
    assert _relerr(net.blob("in0", diff=True), dx_ref) < BTOL, "dx"
    assert _relerr(net.param(0, diff=True), dw_ref) < BTOL, "dW"


def test_conv_bf16_grouped():
    # AlexNet-style grouped conv under bf16 compute (group GEMM slices)
    N, C, H, W, Co, k, p, grp = 2, 8, 14, 14, 16, 5, 2, 2
    x = rng.standard_normal((N, C, H, W)).astype(np.float32)
    w = (rng.standard_normal((Co, C // grp, k, k)) * 0.2).astype(np.float32)
    body = f"""layer {{ name: "conv" type: "Convolution" bottom: "in0"
  top: "out" convolution_param {{ num_output: {Co} kernel_size: {k}
  pad: {p} group: {grp} bias_term: false }} }}"""
    y_ref = orc.conv_fwd(x, w, None, pad=(p, p), stride=(1, 1), group=grp)
    dy = rng.standard_normal(y_ref.shape).astype(np.float32)
    net, y = run_layer("gpu", [(N, C, H, W)], body, [x], params=[w],
                       top_diff=dy)
    assert _relerr(y, y_ref) < BTOL
    dx_ref, dw_ref, _ = orc.conv_bwd(x, w, dy, pad=(p, p), stride=(1, 1),
                                     group=grp, want_db=False)
    assert _relerr(net.blob("in0", diff=True), dx_ref) < BTOL
    assert _relerr(net.param(0, diff=True), dw_ref) < BTOL


@pytest.mark.parametrize("shape", [(64, 64, 64), (130, 1000, 2048),
                                   (64, 147, 40000)])  # last: split-K
def test_ip_bf16(shape):
    M, Nout, K = shape
    x = rng.standard_normal((M, K)).astype(np.float32)
    w = (rng.standard_normal((Nout, K)) * 0.05).astype(np.float32)
    body = f"""layer {{ name: "ip" type: "InnerProduct" bottom: "in0"
  top: "out" inner_product_param {{ num_output: {Nout}
  bias_term: false }} }}"""
    net, y = run_layer("gpu", [(M, K)], body, [x], params=[w])
    y_ref = x.astype(np.float64) @ w.T.astype(np.float64)
    assert _relerr(y, y_ref) < BTOL


def test_resnet50_bf16_step_loss():
    # 2 training steps in bf16 vs f32: same seeds and synthetic stream —
    # the loss trajectory must track within the mixed-precision class
    import os
    import subprocess
    import sys
    from engine_util import REPO

    gen = os.path.join(REPO, "models", "generated",
                       "resnet50_solver.prototxt")
    if not os.path.exists(gen):
        subprocess.check_call([sys.executable,
                               os.path.join(REPO, "models",
                                            "gen_models.py")])
    losses = {}
    for mode in ("f32", "bf16"):
        ca.set_compute(mode)
        ca.set_synthetic_shape(3, 224, 224, 1000)
        ca.set_random_seed(77)
        s = ca.Solver(path=gen, batch_override=4)
        s.step(2)
        losses[mode] = s.loss()
    assert np.isfinite(losses["bf16"])
    assert abs(losses["bf16"] - losses["f32"]) < 0.1 * max(
        1.0, abs(losses["f32"])), losses
