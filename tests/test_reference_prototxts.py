"""Drop-in proof: the reference's OWN untouched model/solver prototxts
(from the read-only /root/reference mount) parse and train on this engine.
Nothing from the reference is copied into the repo — these tests read the
mount directly and skip where it is absent (e.g. on the GPU box; they are
CPU tests and the driver's CPU suite runs where the mount exists).
"""
import os

import numpy as np
import pytest

from engine_util import net_from_text
import caffe_amd as ca

REF = "/root/reference"

needs_ref = pytest.mark.skipif(not os.path.isdir(REF),
                               reason="/root/reference not mounted")


def train_steps(net_path, shape, classes, steps=2, batch=2):
    ca.set_mode("cpu")
    ca.set_synthetic_shape(*shape, classes)
    solver = ca.Solver(text=f"""net: "{net_path}"
base_lr: 0.01
lr_policy: "fixed"
momentum: 0.9
weight_decay: 0.0005
random_seed: 3
snapshot_after_train: false
max_iter: {steps}
""", batch_override=batch)
    solver.step(steps)
    loss = solver.loss()
    assert np.isfinite(loss) and loss > 0, loss
    return loss


@needs_ref
def test_reference_resnet50_train_val():
    train_steps(os.path.join(REF, "models/resnet50/train_val.prototxt"),
                (3, 224, 224), 1000)


@needs_ref
def test_reference_alexnet_train_val():
    train_steps(
        os.path.join(REF, "models/bvlc_alexnet/train_val.prototxt"),
        (3, 227, 227), 1000)


@needs_ref
def test_reference_googlenet_train_val():
    train_steps(
        os.path.join(REF, "models/bvlc_googlenet/train_val.prototxt"),
        (3, 224, 224), 1000)


@needs_ref
@pytest.mark.parametrize("name,shape,classes", [
    ("alexnet_bn", (3, 227, 227), 1000),     # BN variant (engine: CAFFE
                                             # subfield tolerated)
    ("alexnet_owt", (3, 224, 224), 1000),    # one-weird-trick AlexNet
    ("bvlc_reference_caffenet", (3, 227, 227), 1000),  # LRN CaffeNet
    ("resnet18", (3, 224, 224), 1000),       # msra filler
    ("vgg16", (3, 224, 224), 1000),
    ("inception_v2", (3, 224, 224), 1000),   # standalone Softmax aux head
    ("inception_v3", (3, 299, 299), 1000),
])
def test_reference_extra_model_families(name, shape, classes):
    train_steps(os.path.join(REF, "models", name, "train_val.prototxt"),
                shape, classes, steps=1)


@needs_ref
def test_reference_cifar10_nv():
    train_steps(os.path.join(
        REF, "models/cifar10_nv/cifar10_nv_train_test.prototxt"),
        (3, 28, 28), 10, steps=1, batch=4)


@needs_ref
def test_reference_resnet50_solver_file():
    # the reference's own solver.prototxt verbatim (poly policy, comments,
    # relative net path — resolved from the reference root as cwd; nothing
    # is written: snapshot interval 2.5M is never reached and the python
    # API path does not snapshot after train)
    ca.set_mode("cpu")
    ca.set_synthetic_shape(3, 224, 224, 1000)
    cwd = os.getcwd()
    os.chdir(REF)
    try:
        solver = ca.Solver(
            path=os.path.join(REF, "models/resnet50/solver.prototxt"),
            batch_override=2)
        solver.step(2)
        loss = solver.loss()
    finally:
        os.chdir(cwd)
    assert np.isfinite(loss) and loss > 0, loss


@needs_ref
@pytest.mark.parametrize("deploy,prob", [
    ("models/bvlc_alexnet/deploy.prototxt", "prob"),
    ("models/bvlc_reference_rcnn_ilsvrc13/deploy.prototxt",
     "fc-rcnn"),  # ends in an InnerProduct score head
])
def test_reference_deploy_inference(deploy, prob):
    # deploy nets: Input layer + Softmax/score head, pure inference
    ca.set_mode("cpu")
    net = ca.Net.from_file(os.path.join(REF, deploy), phase=1)
    rng = np.random.default_rng(11)
    shape = net.blob_shape("data")
    net.set_blob("data", rng.standard_normal(shape).astype(np.float32))
    net.forward()
    out = net.blob(prob)
    assert np.all(np.isfinite(out))
    if prob == "prob":  # softmax head sums to 1 per image
        s = out.reshape(out.shape[0], -1).sum(1)
        assert np.allclose(s, 1.0, atol=1e-4), s


@needs_ref
def test_reference_lenet_solver_file():
    # the official MNIST solver verbatim (inv policy, comments, relative
    # net path; BASELINE configs[0] is the LeNet CPU case)
    ca.set_mode("cpu")
    ca.set_synthetic_shape(1, 28, 28, 10)
    cwd = os.getcwd()
    os.chdir(REF)
    try:
        solver = ca.Solver(
            path=os.path.join(REF, "examples/mnist/lenet_solver.prototxt"),
            batch_override=8)
        solver.step(3)
        loss = solver.loss()
    finally:
        os.chdir(cwd)
    assert np.isfinite(loss) and loss > 0, loss


@needs_ref
def test_reference_cifar10_nv_solver_file():
    # inline-comment-heavy solver with trailing comments after values
    ca.set_mode("cpu")
    ca.set_synthetic_shape(3, 28, 28, 10)
    cwd = os.getcwd()
    os.chdir(REF)
    try:
        solver = ca.Solver(
            path=os.path.join(
                REF, "models/cifar10_nv/cifar10_nv_solver.prototxt"),
            batch_override=4)
        solver.step(2)
        loss = solver.loss()
    finally:
        os.chdir(cwd)
    assert np.isfinite(loss) and loss > 0, loss


@needs_ref
def test_reference_lenet_train_test():
    train_steps(
        os.path.join(REF, "examples/mnist/lenet_train_test.prototxt"),
        (1, 28, 28), 10, batch=8)
