"""Independent decode of the .caffemodel binaryproto: a from-scratch
varint/wire walker here (NOT the engine's codec) parses the snapshot and
must recover every layer name, blob shape and float value.  Pins the
proto2 wire compatibility of the snapshot format (caffe.proto
NetParameter{name=1, layer=100} / LayerParameter{name=1, blobs=7} /
BlobProto{shape=7{dim=1}, data=5}) rather than a round-trip through the
same code that wrote it.
"""
import os
import struct
import tempfile

import numpy as np

import caffe_amd as ca
from engine_util import net_from_text, input_net


def walk(buf):
    """Yield (field_number, wire_type, value_or_bytes) for one message."""
    i = 0
    n = len(buf)
    while i < n:
        key, i = read_varint(buf, i)
        fnum, wt = key >> 3, key & 7
        if wt == 0:
            v, i = read_varint(buf, i)
            yield fnum, wt, v
        elif wt == 2:
            ln, i = read_varint(buf, i)
            yield fnum, wt, buf[i:i + ln]
            i += ln
        elif wt == 5:
            yield fnum, wt, buf[i:i + 4]
            i += 4
        elif wt == 1:
            yield fnum, wt, buf[i:i + 8]
            i += 8
        else:
            raise AssertionError(f"unexpected wire type {wt}")


def read_varint(buf, i):
    shift = 0
    v = 0
    while True:
        b = buf[i]
        i += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, i
        shift += 7


def decode_blob(buf):
    shape = []
    data = b""
    for fnum, wt, v in walk(buf):
        if fnum == 7 and wt == 2:  # BlobShape
            for f2, w2, v2 in walk(v):
                if f2 == 1:
                    if w2 == 2:  # packed dims
                        j = 0
                        while j < len(v2):
                            d, j = read_varint(v2, j)
                            shape.append(d)
                    else:
                        shape.append(v2)
        elif fnum == 5:  # float data
            if wt == 2:  # packed
                data += v
            else:
                data += v  # single fixed32
    arr = np.frombuffer(data, "<f4")
    return shape, arr


def test_solverstate_independent_decode():
    # SolverState { iter=1, learned_net=2, history=3 (BlobProto) } —
    # caffe.proto:303-308, decoded with the same independent walker
    ca.set_mode("cpu")
    solver = ca.Solver(text="""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 13
snapshot_prefix: "SNAP"
net_param {
  name: "s"
  layer { name: "input" type: "Input" top: "in0" top: "in1"
    input_param { shape { dim: 2 dim: 4 } shape { dim: 2 } } }
  layer { name: "ip" type: "InnerProduct" bottom: "in0" top: "fc"
    inner_product_param { num_output: 3
      weight_filler { type: "gaussian" std: 0.2 } } }
  layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
    bottom: "in1" top: "loss" }
}
""".replace("SNAP", tempfile.mkdtemp() + "/s"))
    rng = np.random.default_rng(1)
    solver.net.set_blob("in0", rng.standard_normal((2, 4)).astype(
        np.float32))
    solver.net.set_blob("in1", np.array([0, 2], np.float32))
    solver.step(3)
    import caffe_amd as ca2
    assert ca2._lib.caffe_solver_snapshot(solver._h) == 0
    # find the state file (prefix was a tmpdir)
    import glob
    states = glob.glob(os.path.join(tempfile.gettempdir(), "**",
                                    "s_iter_3.solverstate"),
                       recursive=True)
    assert states, "no solverstate written"
    raw = open(max(states, key=os.path.getmtime), "rb").read()
    it = None
    learned = None
    hist = []
    for fnum, wt, v in walk(raw):
        if fnum == 1 and wt == 0:
            it = v
        elif fnum == 2 and wt == 2:
            learned = v.decode()
        elif fnum == 3 and wt == 2:
            hist.append(decode_blob(v))
    assert it == 3
    assert learned and learned.endswith("s_iter_3.caffemodel")
    # history blobs mirror the learnable params (ip W and b)
    assert len(hist) == solver.net.num_params()
    assert hist[0][1].size in (12, 3)  # W (3x4) or bias depending on order


def test_caffemodel_independent_decode():
    ca.set_mode("cpu")
    ca.set_random_seed(41)
    net = net_from_text(input_net([(2, 3, 6, 6)], """layer { name: "c1"
  type: "Convolution" bottom: "in0" top: "mid" convolution_param {
  num_output: 4 kernel_size: 3 weight_filler { type: "gaussian"
  std: 0.3 } bias_filler { type: "gaussian" std: 0.1 } } }
layer { name: "ip1" type: "InnerProduct" bottom: "mid" top: "out"
  inner_product_param { num_output: 5
  weight_filler { type: "xavier" } } }"""))
    with tempfile.TemporaryDirectory() as tmp:
        path = os.path.join(tmp, "m.caffemodel")
        net.save_weights(path)
        raw = open(path, "rb").read()

    layers = {}  # name -> [(shape, data), ...]
    for fnum, wt, v in walk(raw):
        if fnum == 100 and wt == 2:  # LayerParameter
            name = None
            blobs = []
            for f2, w2, v2 in walk(v):
                if f2 == 1 and w2 == 2:
                    name = v2.decode()
                elif f2 == 7 and w2 == 2:
                    blobs.append(decode_blob(v2))
            if name is not None:
                layers[name] = blobs

    assert "c1" in layers and "ip1" in layers, layers.keys()
    c1w_shape, c1w = layers["c1"][0]
    assert c1w_shape == [4, 3, 3, 3], c1w_shape
    assert np.array_equal(c1w, net.param(
        [i for i in range(net.num_params())
         if net.param_info(i)[:2] == ("c1", 0)][0]))
    c1b_shape, c1b = layers["c1"][1]
    assert c1b_shape == [4]
    ipw_shape, ipw = layers["ip1"][0]
    assert ipw_shape == [5, 4 * 4 * 4], ipw_shape
    assert ipw.size == 5 * 64
