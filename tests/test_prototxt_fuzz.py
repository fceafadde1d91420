"""Prototxt parser robustness: mutated/truncated/garbage inputs must
either parse or raise a clean engine error (CaffeError/RuntimeError) —
never crash the process.  The reference leans on protobuf's own parser;
this engine's from-scratch text-format parser (csrc/prototxt.cpp) gets
the adversarial treatment instead.
"""
import numpy as np
import pytest

import caffe_amd as ca
from engine_util import net_from_text

BASE = """name: "fz"
layer { name: "input" type: "Input" top: "data"
  input_param { shape { dim: 2 dim: 3 dim: 6 dim: 6 } } }
layer { name: "conv" type: "Convolution" bottom: "data" top: "c"
  convolution_param { num_output: 4 kernel_size: 3
    weight_filler { type: "gaussian" std: 0.1 } } }
layer { name: "relu" type: "ReLU" bottom: "c" top: "c" }
"""


def try_net(text):
    ca.set_mode("cpu")
    try:
        net = net_from_text(text)
        net.forward()
    except Exception:
        pass  # a clean raise is acceptable; a crash fails the run


@pytest.mark.parametrize("cut", [0, 1, 17, 64, 130, 200, 280, 350])
def test_truncations(cut):
    try_net(BASE[:cut])


def test_mutations():
    rng = np.random.default_rng(31)
    raw = BASE.encode()
    for _ in range(60):
        b = bytearray(raw)
        for _ in range(int(rng.integers(1, 6))):
            b[int(rng.integers(0, len(b)))] = int(rng.integers(32, 127))
        try_net(b.decode("latin1"))


def test_garbage_and_structural():
    cases = [
        "", "}", "{", "layer {", "layer { } " * 50,
        "name: \"x\"\nlayer { name: \"a\" type: \"NoSuchLayer\" top: \"t\" }",
        "layer { name: \"a\" type: \"Convolution\" bottom: \"missing\" "
        "top: \"t\" convolution_param { num_output: 1 kernel_size: 1 } }",
        "layer { name: \"a\" type: \"Input\" top: \"t\" input_param { "
        "shape { dim: -4 dim: 3 } } }",
        "layer { name: \"a\" type: \"Input\" top: \"t\" input_param { "
        "shape { dim: 999999999999999 } } }",
        BASE.replace("kernel_size: 3", "kernel_size: 0"),
        BASE.replace("kernel_size: 3", "kernel_size: 9"),  # > input + pad
        BASE.replace("num_output: 4", "num_output: -1"),
        BASE + BASE,  # duplicated graph (duplicate layer names)
    ]
    for c in cases:
        try_net(c)
