"""pycaffe-API shim (caffe_amd.pycaffe): scripts written against classic
pycaffe (reference python/caffe/_caffe.cpp + pycaffe.py surface) run
against this engine with only the import line changed.
"""
import os
import tempfile

import numpy as np

import caffe_amd.pycaffe as caffe
from engine_util import REPO


NET = """name: "p"
force_backward: true
layer { name: "input" type: "Input" top: "data" top: "label"
  input_param { shape { dim: 4 dim: 3 dim: 8 dim: 8 } shape { dim: 4 } } }
layer { name: "conv1" type: "Convolution" bottom: "data" top: "c1"
  convolution_param { num_output: 6 kernel_size: 3
    weight_filler { type: "gaussian" std: 0.2 }
    bias_filler { type: "constant" value: 0.1 } } }
layer { name: "ip1" type: "InnerProduct" bottom: "c1" top: "fc"
  inner_product_param { num_output: 5
    weight_filler { type: "xavier" } } }
layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "label"
  top: "loss" }
"""


def write(tmp, name, text):
    p = os.path.join(tmp, name)
    open(p, "w").write(text)
    return p


def test_net_blobs_params_forward_backward():
    caffe.set_mode_cpu()
    caffe.set_random_seed(7)
    with tempfile.TemporaryDirectory() as tmp:
        net = caffe.Net(write(tmp, "n.prototxt", NET), caffe.TRAIN)
    # dict-shaped surfaces
    assert {"data", "label", "c1", "fc", "loss"} <= set(net.blobs)
    assert "conv1" in net.params and len(net.params["conv1"]) == 2
    assert net.params["conv1"][0].shape == (6, 3, 3, 3)
    assert [l.type for l in net.layers][:2] == ["Input", "Convolution"]
    # zero-copy data assignment, classic pycaffe idiom
    rng = np.random.default_rng(0)
    net.blobs["data"].data[...] = rng.standard_normal((4, 3, 8, 8))
    net.blobs["label"].data[...] = [0, 1, 2, 3]
    net.forward()
    loss = float(net.blobs["loss"].data.ravel()[0])
    assert np.isfinite(loss) and loss > 0
    net.backward()
    g = net.params["conv1"][0].diff
    assert g.shape == (6, 3, 3, 3) and np.abs(g).max() > 0
    # mutating weights through the view changes the next forward
    net.params["ip1"][0].data[...] = 0
    net.forward()
    l2 = float(net.blobs["loss"].data.ravel()[0])
    assert abs(l2 - np.log(5)) < 1e-5  # uniform logits -> ln(C)


def test_save_copy_from_roundtrip():
    caffe.set_mode_cpu()
    caffe.set_random_seed(13)
    with tempfile.TemporaryDirectory() as tmp:
        p = write(tmp, "n.prototxt", NET)
        net = caffe.Net(p, caffe.TRAIN)
        w = net.params["conv1"][0].data.copy()
        model = os.path.join(tmp, "m.caffemodel")
        net.save(model)
        caffe.set_random_seed(99)  # different init
        net2 = caffe.Net(p, model, caffe.TRAIN)  # (proto, weights, phase)
        assert np.allclose(net2.params["conv1"][0].data, w)


def test_sgdsolver_step_and_snapshot():
    caffe.set_mode_cpu()
    with tempfile.TemporaryDirectory() as tmp:
        netp = write(tmp, "n.prototxt", NET.replace(
            'type: "Input"', 'type: "Input"').replace(
            "force_backward: true\n", ""))
        solver_txt = f"""net: "{netp}"
base_lr: 0.1
lr_policy: "fixed"
momentum: 0.9
max_iter: 4
random_seed: 3
snapshot_prefix: "{tmp}/s"
"""
        s = caffe.SGDSolver(write(tmp, "s.prototxt", solver_txt))
        rng = np.random.default_rng(5)
        s.net.blobs["data"].data[...] = rng.standard_normal((4, 3, 8, 8))
        s.net.blobs["label"].data[...] = [0, 1, 2, 3]
        l0 = None
        for _ in range(6):
            s.step(1)
            s.net.blobs["data"].data[...] = rng.standard_normal(
                (4, 3, 8, 8)) * 0 + s.net.blobs["data"].data
            l = float(s.net.blobs["loss"].data.ravel()[0])
            if l0 is None:
                l0 = l
        assert s.iter == 6
        assert l < l0  # fixed batch: SGD must reduce the loss
        s.snapshot()
        state = os.path.join(tmp, f"s_iter_{s.iter}.solverstate")
        assert os.path.exists(state)
        s2 = caffe.SGDSolver(os.path.join(tmp, "s.prototxt"))
        s2.restore(state)
        assert s2.iter == 6


import pytest


@pytest.mark.gpu
def test_pycaffe_gpu_solver():
    # the classic pycaffe GPU flow end-to-end on the engine
    caffe.set_device(0)
    caffe.set_mode_gpu()
    caffe.set_random_seed(21)
    with tempfile.TemporaryDirectory() as tmp:
        netp = write(tmp, "n.prototxt", NET.replace(
            "force_backward: true\n", ""))
        s = caffe.SGDSolver(write(tmp, "s.prototxt", f"""net: "{netp}"
base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 3
snapshot_prefix: "{tmp}/s"
"""))
        rng = np.random.default_rng(5)
        s.net.blobs["data"].data[...] = rng.standard_normal((4, 3, 8, 8))
        s.net.blobs["label"].data[...] = [0, 1, 2, 3]
        l0 = None
        for _ in range(8):
            s.step(1)
            l = float(s.net.blobs["loss"].data.ravel()[0])
            l0 = l if l0 is None else l0
        assert np.isfinite(l) and l < l0
    caffe.set_mode_cpu()


def test_sgdsolver_snapshot_restore_roundtrip():
    # solver.snapshot() / solver.restore(state): params + iter survive the
    # round trip through a fresh solver (pycaffe SGDSolver surface)
    caffe.set_mode_cpu()
    with tempfile.TemporaryDirectory() as tmp:
        sol = f"""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 11
snapshot_prefix: "{tmp}/s"
net_param {{ {NET[NET.index('name'):]} }}
"""
        s = caffe.SGDSolver(write(tmp, "s.prototxt", sol))
        s.step(3)
        assert s.iter == 3
        w = s.net.params["conv1"][0].data.copy()
        s.snapshot()
        state = os.path.join(tmp, "s_iter_3.solverstate")
        assert os.path.exists(state)
        s2 = caffe.SGDSolver(write(tmp, "s2.prototxt", sol))
        s2.restore(state)
        assert s2.iter == 3
        assert np.array_equal(s2.net.params["conv1"][0].data, w)
        s2.step(1)  # training continues from the restored state
        assert s2.iter == 4
