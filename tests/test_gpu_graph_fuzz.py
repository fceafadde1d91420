"""GPU graph-fusion fuzz: random graphs built to TRIGGER the GPU-only
graph passes (BN+in-place-ReLU fusion, Eltwise+ReLU fusion, the
BN+residual-add epilogue, conv-epilogue ReLU, in-place BN, strided 1x1
scatter-dgrad, strided implicit views) and compared CPU vs GPU on loss,
input gradient and every param gradient.  The fixed model tests cover
these fusions only in the exact ResNet/GoogLeNet shapes; here the
combinations are randomized (fixed seeds — deterministic).
"""
import numpy as np
import pytest

import caffe_amd as ca
from engine_util import net_from_text

pytestmark = pytest.mark.gpu


def gen_graph(rng):
    n = int(rng.integers(2, 4))
    c = int(rng.integers(3, 6))
    hw = int(rng.choice([8, 9, 12]))
    lines = []
    cur = "in0"
    li = 0
    for _ in range(int(rng.integers(2, 5))):
        kind = rng.choice(["convbnrelu", "residual", "conv_inplace_relu",
                           "strided1x1", "inplace_bn", "pool"])
        li += 1
        nm = f"l{li}"
        top = f"t{li}"
        if kind == "convbnrelu":
            co = int(rng.integers(3, 7))
            lines += [
                f'layer {{ name: "{nm}c" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}" convolution_param {{ '
                f'num_output: {co} kernel_size: 3 pad: 1 '
                f'weight_filler {{ type: "gaussian" std: 0.3 }} }} }}',
                f'layer {{ name: "{nm}b" type: "BatchNorm" '
                f'bottom: "{top}" top: "{top}bn" '
                f'batch_norm_param {{ scale_bias: true }} }}',
                # in-place ReLU directly after BN -> fused epilogue
                f'layer {{ name: "{nm}r" type: "ReLU" '
                f'bottom: "{top}bn" top: "{top}bn" }}']
            cur = f"{top}bn"
        elif kind == "residual":
            co = int(rng.integers(3, 6))
            lines += [
                # two branches from cur: conv+BN vs 1x1 conv; SUM; ReLU
                f'layer {{ name: "{nm}a" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}a" convolution_param {{ '
                f'num_output: {co} kernel_size: 3 pad: 1 '
                f'weight_filler {{ type: "gaussian" std: 0.3 }} }} }}',
                f'layer {{ name: "{nm}ab" type: "BatchNorm" '
                f'bottom: "{top}a" top: "{top}abn" '
                f'batch_norm_param {{ scale_bias: true }} }}',
                f'layer {{ name: "{nm}s" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}s" convolution_param {{ '
                f'num_output: {co} kernel_size: 1 bias_term: false '
                f'weight_filler {{ type: "gaussian" std: 0.3 }} }} }}',
                f'layer {{ name: "{nm}e" type: "Eltwise" '
                f'bottom: "{top}abn" bottom: "{top}s" top: "{top}" '
                f'eltwise_param {{ operation: SUM }} }}',
                f'layer {{ name: "{nm}r" type: "ReLU" '
                f'bottom: "{top}" top: "{top}" }}']
            cur = top
        elif kind == "conv_inplace_relu":
            co = int(rng.integers(3, 6))
            lines += [
                f'layer {{ name: "{nm}c" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}" convolution_param {{ '
                f'num_output: {co} kernel_size: 3 pad: 1 '
                f'weight_filler {{ type: "gaussian" std: 0.3 }} '
                f'bias_filler {{ type: "gaussian" std: 0.1 }} }} }}',
                f'layer {{ name: "{nm}r" type: "ReLU" '
                f'bottom: "{top}" top: "{top}" }}']
            cur = top
        elif kind == "strided1x1":
            co = int(rng.integers(3, 6))
            lines += [
                f'layer {{ name: "{nm}c" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}" convolution_param {{ '
                f'num_output: {co} kernel_size: 1 stride: 2 '
                f'bias_term: false '
                f'weight_filler {{ type: "gaussian" std: 0.3 }} }} }}']
            cur = top
        elif kind == "inplace_bn":
            lines += [
                f'layer {{ name: "{nm}b" type: "BatchNorm" '
                f'bottom: "{cur}" top: "{cur}" '
                f'batch_norm_param {{ scale_bias: true }} }}']
        else:
            lines += [
                f'layer {{ name: "{nm}p" type: "Pooling" bottom: "{cur}" '
                f'top: "{top}" pooling_param {{ pool: MAX kernel_size: 2 '
                f'stride: 1 }} }}']
            cur = top
    lines += [
        f'layer {{ name: "ip" type: "InnerProduct" bottom: "{cur}" '
        f'top: "fc" inner_product_param {{ num_output: 4 '
        f'weight_filler {{ type: "gaussian" std: 0.2 }} }} }}',
        f'layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc" '
        f'bottom: "in1" top: "loss" }}']
    head = (f'name: "fz"\nforce_backward: true\n'
            f'layer {{ name: "input" type: "Input" top: "in0" '
            f'top: "in1" input_param {{ '
            f'shape {{ dim: {n} dim: {c} dim: {hw} dim: {hw} }} '
            f'shape {{ dim: {n} }} }} }}\n')
    return head + "\n".join(lines), (n, c, hw)


@pytest.mark.parametrize("dtype", ["f32", "bf16"])
@pytest.mark.parametrize("seed", [5, 19, 42, 77, 101, 137])
def test_gpu_graph_fusion_parity(seed, dtype):
    # tol: exact-class for f32, the bf16 mixed-precision class otherwise
    tol = 1e-3 if dtype == "f32" else 4e-2
    rng = np.random.default_rng(seed)
    text, (n, c, hw) = gen_graph(rng)
    x = rng.standard_normal((n, c, hw, hw)).astype(np.float32)
    labels = rng.integers(0, 4, n).astype(np.float32)
    results = {}
    try:
        for mode in ("cpu", "gpu"):
            ca.set_mode(mode)
            if mode == "gpu":
                ca.set_compute(dtype)
            ca.set_random_seed(1000 + seed)
            net = net_from_text(text)
            net.set_blob("in0", x)
            net.set_blob("in1", labels)
            net.forward()
            net.backward()
            results[mode] = (
                float(np.asarray(net.blob("loss")).ravel()[0]),
                np.asarray(net.blob("in0", diff=True)).copy(),
                [np.asarray(net.param(i, diff=True)).copy()
                 for i in range(net.num_params())])
    finally:
        ca.set_compute("f32")
    lc, dc, pc = results["cpu"]
    lg, dg, pg = results["gpu"]
    assert abs(lg - lc) < tol * max(1.0, abs(lc)), (lc, lg)
    scale = max(np.abs(dc).max(), 1e-3)
    assert np.abs(dg - dc).max() < tol * scale
    for i, (a, b) in enumerate(zip(pc, pg)):
        s = max(np.abs(a).max(), 1e-3)
        assert np.abs(b - a).max() < tol * s, f"param {i}"
