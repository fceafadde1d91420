"""Shared harness for engine-vs-oracle parity tests (CPU and GPU)."""
import os
import sys
import tempfile

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))
sys.path.insert(0, REPO)

import caffe_amd as ca  # noqa: E402

TOL = 1e-4


def relerr(a, b):
    # ravel both sides: comparisons are elementwise in memory order, and a
    # shape mismatch must be an error, never a silent numpy broadcast
    a = np.asarray(a, np.float64).ravel()
    b = np.asarray(b, np.float64).ravel()
    assert a.size == b.size, (a.size, b.size)
    denom = max(np.abs(b).max(), 1e-8)
    return np.abs(a - b).max() / denom


def net_from_text(text, phase=0):
    f = tempfile.NamedTemporaryFile("w", suffix=".prototxt", delete=False)
    f.write(text)
    f.close()
    return ca.Net.from_file(f.name, phase=phase)


def input_net(shapes, layer_body, extra=""):
    """Single-layer net: Input tops -> one layer under test."""
    shape_txt = "".join(
        "    shape { " + " ".join(f"dim: {d}" for d in s) + " }\n"
        for s in shapes)
    tops = "".join(f'  top: "in{i}"\n' for i in range(len(shapes)))
    return f"""name: "t"
force_backward: true
layer {{
  name: "input"
  type: "Input"
{tops}  input_param {{
{shape_txt}  }}
}}
{layer_body}
{extra}
"""


def run_layer(mode, shapes, layer_body, inputs, params=None, top="out",
              top_diff=None, phase=0):
    """Build the net, set inputs/params, forward (+backward when top_diff
    given).  Returns (net, out)."""
    ca.set_mode(mode)
    net = net_from_text(input_net(shapes, layer_body), phase=phase)
    for i, x in enumerate(inputs):
        net.set_blob(f"in{i}", x)
    if params is not None:
        for idx, p in enumerate(params):
            if p is not None:
                net.set_param(idx, p.ravel())
    net.forward()
    out = net.blob(top)
    if top_diff is not None:
        net.set_blob(top, top_diff, diff=True)
        net.backward()
    return net, out
