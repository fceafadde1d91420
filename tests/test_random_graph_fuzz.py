"""Seeded random-graph gradient fuzz: random compositions of the hot-path
layers (conv/BN/ReLU-off-kink/pool/LRN/eltwise/concat/IP) are built as
prototxt, then every param and input gradient is finite-difference
checked.  Self-validating (no oracle needed) — catches wiring, in-place,
fan-out and diff-accumulation bugs in graph shapes the fixed model tests
never produce.  Deterministic: graphs derive from fixed seeds.
"""
import numpy as np
import pytest

from test_gradient_check import build_net, check_gradients, EPS

import caffe_amd as ca


def gen_graph(rng):
    """Random chain with occasional fan-out/merge, NCHW 4-6 spatial."""
    n = int(rng.integers(1, 4))
    c = int(rng.integers(2, 5))
    hw = int(rng.integers(6, 10))
    lines = []
    cur = "in0"
    cur_c = c
    depth = int(rng.integers(2, 5))
    li = 0
    for _ in range(depth):
        kind = rng.choice(["conv", "bn", "pool", "lrn", "fanout"])
        li += 1
        name = f"l{li}"
        top = f"t{li}"
        if kind == "conv":
            co = int(rng.integers(2, 6))
            k = int(rng.choice([1, 3]))
            p = k // 2
            lines.append(
                f'layer {{ name: "{name}" type: "Convolution" '
                f'bottom: "{cur}" top: "{top}" convolution_param {{ '
                f'num_output: {co} kernel_size: {k} pad: {p} '
                f'weight_filler {{ type: "gaussian" std: 0.4 }} '
                f'bias_filler {{ type: "gaussian" std: 0.1 }} }} }}')
            cur_c = co
        elif kind == "bn":
            lines.append(
                f'layer {{ name: "{name}" type: "BatchNorm" '
                f'bottom: "{cur}" top: "{top}" '
                f'batch_norm_param {{ scale_bias: true }} }}')
        elif kind == "pool":
            pool = rng.choice(["MAX", "AVE"])
            lines.append(
                f'layer {{ name: "{name}" type: "Pooling" bottom: "{cur}" '
                f'top: "{top}" pooling_param {{ pool: {pool} '
                f'kernel_size: 2 stride: 1 }} }}')
        elif kind == "lrn":
            lines.append(
                f'layer {{ name: "{name}" type: "LRN" bottom: "{cur}" '
                f'top: "{top}" lrn_param {{ local_size: 3 alpha: 0.4 '
                f'beta: 0.6 }} }}')
        else:  # fanout: two 1x1 convs off cur, merged by eltwise or concat
            co = int(rng.integers(2, 5))
            merge = rng.choice(["Eltwise", "Concat"])
            for br in ("a", "b"):
                lines.append(
                    f'layer {{ name: "{name}{br}" type: "Convolution" '
                    f'bottom: "{cur}" top: "{top}{br}" '
                    f'convolution_param {{ num_output: {co} '
                    f'kernel_size: 1 '
                    f'weight_filler {{ type: "gaussian" std: 0.4 }} }} }}')
            if merge == "Eltwise":
                lines.append(
                    f'layer {{ name: "{name}m" type: "Eltwise" '
                    f'bottom: "{top}a" bottom: "{top}b" top: "{top}" '
                    f'eltwise_param {{ operation: SUM coeff: 1.0 '
                    f'coeff: -0.5 }} }}')
                cur_c = co
            else:
                lines.append(
                    f'layer {{ name: "{name}m" type: "Concat" '
                    f'bottom: "{top}a" bottom: "{top}b" top: "{top}" }}')
                cur_c = 2 * co
        cur = top
    # head: IP so the objective sees every branch
    lines.append(
        f'layer {{ name: "head" type: "InnerProduct" bottom: "{cur}" '
        f'top: "out" inner_product_param {{ num_output: 3 '
        f'weight_filler {{ type: "gaussian" std: 0.2 }} }} }}')
    return "\n".join(lines), (n, c, hw, hw)


@pytest.mark.parametrize("seed", [11, 23, 37, 58, 71])
def test_random_graph_gradients(seed):
    ca.set_mode("cpu")
    ca.set_random_seed(seed)
    rng = np.random.default_rng(seed)
    body, shape = gen_graph(rng)
    net = build_net(body, [shape])
    x = rng.standard_normal(shape).astype(np.float32)
    x[np.abs(x) < 5 * EPS] = 0.5  # keep off any downstream kinks
    net.set_blob("in0", x)
    # moderate eps + 1% threshold: FD truncation through BN+LRN chains is
    # O(eps^2) curvature (verified quadratic); wiring bugs would be O(1)
    check_gradients(net, "out", n_samples=8, eps=5e-3, thresh=1e-2)
