"""Edge-shape pins for graph layers the model suites only hit in one
configuration: N-way Concat (GoogLeNet inceptions are 4-way), true
global pooling on odd spatial sizes, Split-created fan-out diffs.
"""
import numpy as np

from engine_util import relerr, run_layer


def test_concat_four_way_forward_backward():
    shapes = [(2, c, 3, 3) for c in (1, 2, 3, 4)]
    rng = np.random.default_rng(1)
    xs = [rng.standard_normal(s).astype(np.float32) for s in shapes]
    dy = rng.standard_normal((2, 10, 3, 3)).astype(np.float32)
    body = """layer { name: "cc" type: "Concat" bottom: "in0"
  bottom: "in1" bottom: "in2" bottom: "in3" top: "out" }"""
    net, out = run_layer("cpu", shapes, body, xs, top_diff=dy)
    assert np.array_equal(out, np.concatenate(xs, axis=1))
    off = 0
    for i, s in enumerate(shapes):
        c = s[1]
        assert np.array_equal(net.blob(f"in{i}", diff=True),
                              dy[:, off:off + c]), i
        off += c


def test_global_pooling_odd_spatial():
    # global_pooling: true — kernel spans the whole (odd) map
    shape = (2, 3, 7, 5)
    rng = np.random.default_rng(2)
    x = rng.standard_normal(shape).astype(np.float32)
    for pool, ref in (("AVE", x.mean(axis=(2, 3))),
                      ("MAX", x.max(axis=(2, 3)))):
        body = f"""layer {{ name: "p" type: "Pooling" bottom: "in0"
  top: "out" pooling_param {{ pool: {pool} global_pooling: true }} }}"""
        _, out = run_layer("cpu", [shape], body, [x])
        assert out.shape[:2] == (2, 3) and out.size == 6
        assert relerr(out.reshape(2, 3), ref) < 1e-5, pool


def test_split_fanout_diff_sum():
    # one top consumed twice: diffs must sum (insert_splits); verified
    # against the closed form dx = dy_a*2 (two identical IP consumers
    # sharing weights would differ — use distinct weights and compose)
    rng = np.random.default_rng(3)
    x = rng.standard_normal((2, 4)).astype(np.float32)
    w1 = rng.standard_normal((3, 4)).astype(np.float32)
    w2 = rng.standard_normal((3, 4)).astype(np.float32)
    dy = rng.standard_normal((2, 3)).astype(np.float32)
    body = """layer { name: "a" type: "InnerProduct" bottom: "in0"
  top: "ta" inner_product_param { num_output: 3 bias_term: false } }
layer { name: "b" type: "InnerProduct" bottom: "in0" top: "tb"
  inner_product_param { num_output: 3 bias_term: false } }
layer { name: "e" type: "Eltwise" bottom: "ta" bottom: "tb" top: "out"
  eltwise_param { operation: SUM } }"""
    net, _ = run_layer("cpu", [(2, 4)], body, [x], params=[w1, w2],
                       top_diff=dy)
    expect = dy @ w1 + dy @ w2
    assert relerr(net.blob("in0", diff=True), expect) < 1e-5
