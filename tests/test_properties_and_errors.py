"""Reference-test property restatements + error-path behavior.

Properties are the reference's own test assertions restated as data-free
invariants (test_batch_norm_layer.cpp mean/var checks, softmax
normalization); error paths check the glog-CHECK-to-status-code
conversion at the C ABI (SURVEY §8b error convention).
"""
import numpy as np
import pytest

import caffe_amd as ca
from engine_util import net_from_text, input_net, run_layer


def test_bn_output_normalized():
    # test_batch_norm_layer.cpp: per-channel mean ~ 0, variance ~ 1
    rng = np.random.default_rng(0)
    x = (rng.standard_normal((6, 4, 7, 7)) * 3 + 5).astype(np.float32)
    body = """layer { name: "bn" type: "BatchNorm" bottom: "in0"
  top: "out" }"""
    _, out = run_layer("cpu", [(6, 4, 7, 7)], body, [x])
    per_c = out.transpose(1, 0, 2, 3).reshape(4, -1)
    assert np.abs(per_c.mean(1)).max() < 1e-4
    assert np.abs(per_c.var(1) - 1).max() < 1e-2


def test_softmax_rows_sum_to_one():
    rng = np.random.default_rng(1)
    x = (rng.standard_normal((5, 11)) * 10).astype(np.float32)
    body = """layer { name: "s" type: "Softmax" bottom: "in0"
  top: "out" }"""
    _, out = run_layer("cpu", [(5, 11)], body, [x])
    assert np.allclose(out.sum(1), 1.0, atol=1e-5)
    assert out.min() >= 0


def test_dropout_test_phase_identity():
    rng = np.random.default_rng(2)
    x = rng.standard_normal((3, 4, 5, 5)).astype(np.float32)
    body = """layer { name: "d" type: "Dropout" bottom: "in0" top: "out"
  dropout_param { dropout_ratio: 0.5 } }"""
    _, out = run_layer("cpu", [(3, 4, 5, 5)], body, [x], phase=1)
    assert np.array_equal(out, x)


def test_unknown_layer_type_raises():
    ca.set_mode("cpu")
    with pytest.raises(Exception) as e:
        net_from_text(input_net([(1, 2)], """layer { name: "x"
  type: "FancyUnsupportedLayer" bottom: "in0" top: "out" }"""))
    assert "FancyUnsupportedLayer" in str(e.value)


def test_missing_solver_file_raises():
    ca.set_mode("cpu")
    with pytest.raises(Exception):
        ca.Solver(path="/nonexistent/solver.prototxt")


def test_shape_mismatch_raises():
    # Eltwise with mismatched bottoms must fail loudly, not corrupt
    ca.set_mode("cpu")
    with pytest.raises(Exception):
        net_from_text(input_net(
            [(2, 3, 4, 4), (2, 3, 5, 5)],
            """layer { name: "e" type: "Eltwise" bottom: "in0"
  bottom: "in1" top: "out" }"""))


def test_malformed_prototxt_raises():
    ca.set_mode("cpu")
    with pytest.raises(Exception):
        net_from_text("layer { name: \"x\" type: ")


def test_truncation_fuzz_never_crashes():
    # every prefix-truncation of a valid net must either parse+build or
    # raise a clean error through the C ABI — never take the process down
    ca.set_mode("cpu")
    full = input_net([(1, 2, 4, 4)], """layer { name: "c"
  type: "Convolution" bottom: "in0" top: "mid" convolution_param {
  num_output: 2 kernel_size: 3 pad: 1 } }
layer { name: "r" type: "ReLU" bottom: "mid" top: "out" }""")
    rng = np.random.default_rng(5)
    for cut in sorted(rng.integers(1, len(full), size=40).tolist()):
        try:
            net_from_text(full[:cut])
        except Exception:
            pass  # clean failure is the contract


def test_error_names_the_layer():
    # round-2 diagnostics: a CHECK failing deep in blob plumbing carries a
    # "[in layer '<name>' (<type>) <stage>]" breadcrumb (the round-1
    # review hit a bare "CHECK failed: (d) >= (0)" with no context)
    import caffe_amd as ca
    from engine_util import input_net, net_from_text
    ca.set_mode("cpu")
    body = """layer { name: "badconv" type: "Convolution" bottom: "in0"
  top: "out" convolution_param { num_output: 4 kernel_size: 9 } }"""
    try:
        net_from_text(input_net([(1, 3, 5, 5)], body))
        raise AssertionError("expected a shape failure")
    except ca.CaffeError as e:
        assert "badconv" in str(e), str(e)


def test_unimplemented_options_fail_loudly():
    # silently mis-computing an unsupported prototxt option is worse than
    # aborting: axis-2 softmax, transposed IP weights, axis-0 concat
    import caffe_amd as ca
    from engine_util import input_net, net_from_text
    ca.set_mode("cpu")
    cases = [
        """layer { name: "s" type: "Softmax" bottom: "in0" top: "out"
  softmax_param { axis: 2 } }""",
        """layer { name: "ip" type: "InnerProduct" bottom: "in0"
  top: "out" inner_product_param { num_output: 3 transpose: true
    weight_filler { type: "xavier" } } }""",
        """layer { name: "ip" type: "InnerProduct" bottom: "in0"
  top: "out" inner_product_param { num_output: 3 axis: 2
    weight_filler { type: "xavier" } } }""",
    ]
    for body in cases:
        try:
            net_from_text(input_net([(2, 3, 4, 4)], body))
            raise AssertionError("expected rejection: " + body[:40])
        except ca.CaffeError as e:
            assert "implement" in str(e), str(e)
