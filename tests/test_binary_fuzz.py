"""Binary-format robustness: mutated/truncated .caffemodel and
.solverstate files must load cleanly or raise a clean engine error —
never crash.  Exercises the from-scratch protobuf wire reader
(csrc/proto_wire) against adversarial input (the reference trusts
protobuf's own parser here).  Verified crash-free in subprocess
isolation first; kept in-process with fixed seeds for CI speed.
"""
import os

import numpy as np
import pytest

import caffe_amd as ca
from engine_util import net_from_text

NET = """name: "t"
layer { name: "input" type: "Input" top: "data"
  input_param { shape { dim: 2 dim: 3 dim: 6 dim: 6 } } }
layer { name: "conv" type: "Convolution" bottom: "data" top: "c"
  convolution_param { num_output: 4 kernel_size: 3
    weight_filler { type: "gaussian" std: 0.2 } } }
layer { name: "ip" type: "InnerProduct" bottom: "c" top: "fc"
  inner_product_param { num_output: 3
    weight_filler { type: "xavier" } } }
"""


def mutate(raw, rng, t):
    b = bytearray(raw)
    mode = t % 3
    if mode == 0:      # random byte flips
        for _ in range(int(rng.integers(1, 8))):
            b[int(rng.integers(0, len(b)))] = int(rng.integers(0, 256))
    elif mode == 1:    # truncation
        b = b[:int(rng.integers(0, len(b)))]
    else:              # header-area flips (field tags / varints)
        for _ in range(int(rng.integers(1, 5))):
            b[int(rng.integers(0, min(200, len(b))))] = \
                int(rng.integers(0, 256))
    return bytes(b)


def test_caffemodel_mutations(tmp_path):
    ca.set_mode("cpu")
    net = net_from_text(NET)
    mdl = str(tmp_path / "m.caffemodel")
    net.save_weights(mdl)
    raw = open(mdl, "rb").read()
    rng = np.random.default_rng(5)
    bad = str(tmp_path / "bad.caffemodel")
    for t in range(60):
        open(bad, "wb").write(mutate(raw, rng, t))
        try:
            net.load_weights(bad)
            net.forward()
        except Exception:
            pass  # clean raise OK; a crash fails the whole run


def test_solverstate_mutations(tmp_path):
    ca.set_mode("cpu")
    text = f"""base_lr: 0.01
lr_policy: "fixed"
momentum: 0.9
snapshot_prefix: "{tmp_path}/s"
net_param {{ name: "n"
layer {{ name: "input" type: "Input" top: "data" top: "label"
  input_param {{ shape {{ dim: 2 dim: 3 dim: 6 dim: 6 }}
    shape {{ dim: 2 }} }} }}
layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
  inner_product_param {{ num_output: 3
    weight_filler {{ type: "xavier" }} }} }}
layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
  bottom: "label" top: "loss" }} }}
"""
    s = ca.Solver(text=text)
    s.step(2)
    assert ca._lib.caffe_solver_snapshot(s._h) == 0
    st = str(tmp_path / "s_iter_2.solverstate")
    raw = open(st, "rb").read()
    rng = np.random.default_rng(13)
    bad = str(tmp_path / "bad.solverstate")
    for t in range(45):
        open(bad, "wb").write(mutate(raw, rng, t))
        s2 = ca.Solver(text=text)
        try:
            ca._ck(ca._lib.caffe_solver_restore(s2._h, bad.encode()))
            s2.step(1)
        except Exception:
            pass


def test_lmdb_mutations(tmp_path):
    # corrupted LMDB databases: the mmap B+tree walker must raise or read
    # garbage-but-bounded, never fault (page offsets are CHECKed against
    # the map)
    import subprocess
    import sys

    from engine_util import REPO
    sys.path.insert(0, os.path.join(REPO, "tools"))
    from make_lmdb import make_lmdb

    db = str(tmp_path / "db")
    make_lmdb(db, 24, 3, 8, 8, 3)
    raw = open(os.path.join(db, "data.mdb"), "rb").read()
    rng = np.random.default_rng(23)
    ca.set_mode("cpu")
    for t in range(50):
        b = bytearray(raw)
        mode = t % 3
        if mode == 0:
            for _ in range(int(rng.integers(1, 10))):
                b[int(rng.integers(0, len(b)))] = int(rng.integers(0, 256))
        elif mode == 1:
            b = b[:int(rng.integers(4096, len(b)))]  # keep meta pages
        else:  # B+tree area flips
            for _ in range(int(rng.integers(1, 6))):
                b[int(rng.integers(8192, len(b)))] = int(rng.integers(0, 256))
        bd = tmp_path / f"bad{t}"
        bd.mkdir(exist_ok=True)
        (bd / "data.mdb").write_bytes(bytes(b))
        try:
            net = net_from_text(f"""name: "t"
layer {{ name: "data" type: "Data" top: "data" top: "label"
  data_param {{ source: "{bd}" batch_size: 4 }} }}
""")
            net.forward()
        except Exception:
            pass
