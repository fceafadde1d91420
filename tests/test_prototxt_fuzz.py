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


def test_wrong_arity_and_shape_mismatch():
    # arity violations must hit the blob-count contract; shape mismatches
    # must hit layer CHECKs — all as clean raises
    cases = [
        # Eltwise with one bottom
        'layer { name: "i" type: "Input" top: "a" input_param { shape { '
        'dim: 2 dim: 3 } } }\n'
        'layer { name: "e" type: "Eltwise" bottom: "a" top: "t" }',
        # SoftmaxWithLoss with one bottom
        'layer { name: "i" type: "Input" top: "a" input_param { shape { '
        'dim: 2 dim: 3 } } }\n'
        'layer { name: "l" type: "SoftmaxWithLoss" bottom: "a" top: "t" }',
        # Accuracy with one bottom
        'layer { name: "i" type: "Input" top: "a" input_param { shape { '
        'dim: 2 dim: 3 } } }\n'
        'layer { name: "l" type: "Accuracy" bottom: "a" top: "t" }',
        # ReLU with no bottom at all (the original segfault)
        'layer { name: "r" type: "ReLU" top: "t" }',
        # Conv with two tops
        'layer { name: "i" type: "Input" top: "a" input_param { shape { '
        'dim: 2 dim: 3 dim: 6 dim: 6 } } }\n'
        'layer { name: "c" type: "Convolution" bottom: "a" top: "t" '
        'top: "u" convolution_param { num_output: 2 kernel_size: 1 } }',
        # Eltwise shape mismatch
        'layer { name: "i" type: "Input" top: "a" top: "b" input_param { '
        'shape { dim: 2 dim: 3 } shape { dim: 2 dim: 5 } } }\n'
        'layer { name: "e" type: "Eltwise" bottom: "a" bottom: "b" '
        'top: "t" }',
        # Concat non-axis shape mismatch
        'layer { name: "i" type: "Input" top: "a" top: "b" input_param { '
        'shape { dim: 2 dim: 3 dim: 4 dim: 4 } '
        'shape { dim: 2 dim: 3 dim: 5 dim: 4 } } }\n'
        'layer { name: "c" type: "Concat" bottom: "a" bottom: "b" '
        'top: "t" }',
        # label shape mismatch for the loss
        'layer { name: "i" type: "Input" top: "a" top: "lab" input_param '
        '{ shape { dim: 4 dim: 3 } shape { dim: 7 } } }\n'
        'layer { name: "l" type: "SoftmaxWithLoss" bottom: "a" '
        'bottom: "lab" top: "t" }',
    ]
    for c in cases:
        with pytest.raises(Exception):
            net = net_from_text('name: "t"\n' + c)
            net.forward()


SOLVER = """base_lr: 0.01
lr_policy: "step"
gamma: 0.1
stepsize: 10
momentum: 0.9
weight_decay: 0.0005
net_param { name: "n"
layer { name: "input" type: "Input" top: "data" top: "label"
  input_param { shape { dim: 2 dim: 3 dim: 6 dim: 6 } shape { dim: 2 } } }
layer { name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
  inner_product_param { num_output: 3
    weight_filler { type: "xavier" } } }
layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "label"
  top: "loss" } }
"""


def try_solver(text):
    ca.set_mode("cpu")
    try:
        s = ca.Solver(text=text)
        s.step(1)
    except Exception:
        pass


@pytest.mark.parametrize("cut", [0, 30, 90, 170, 300, 460])
def test_solver_truncations(cut):
    try_solver(SOLVER[:cut])


def test_solver_mutations():
    rng = np.random.default_rng(97)
    raw = SOLVER.encode()
    for _ in range(50):
        b = bytearray(raw)
        for _ in range(int(rng.integers(1, 6))):
            b[int(rng.integers(0, len(b)))] = int(rng.integers(32, 127))
        try_solver(b.decode("latin1"))
