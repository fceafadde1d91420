"""Pin the oracle (oracle/oracle.c) against golden vectors.

Two anchors:
 1. torch-CPU-generated fixtures under tests/golden/ (generator committed:
    tests/golden/gen_golden.py) — the reference itself cannot be compiled in
    this container.
 2. Known-answer constants restated as *data* from the reference's own tests
    (src/caffe/test/test_pooling_layer.cpp:52-125 square max-pool case).
"""
import os

import numpy as np
import pytest

from oracle import oracle as orc

GOLD = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def load(name):
    return dict(np.load(os.path.join(GOLD, name + ".npz")))


def relerr(a, b):
    a = np.asarray(a, np.float64)
    b = np.asarray(b, np.float64)
    denom = max(np.abs(b).max(), 1e-8)
    return np.abs(a - b).max() / denom


TOL = 1e-4  # the reference's own EXPECT_NEAR class (test_convolution_layer.cpp:247)


@pytest.mark.parametrize("i", range(6))
def test_conv_golden(i):
    g = load(f"conv{i}")
    N, C, H, W, Co, k, s, p, grp = [int(v) for v in g["meta"]]
    y = orc.conv_fwd(g["x"], g["w"], g["b"], pad=(p, p), stride=(s, s),
                     group=grp)
    assert relerr(y, g["y"]) < TOL
    dx, dw, db = orc.conv_bwd(g["x"], g["w"], g["dy"], pad=(p, p),
                              stride=(s, s), group=grp, want_db=True)
    assert relerr(dx, g["dx"]) < TOL
    assert relerr(dw, g["dw"]) < TOL
    assert relerr(db, g["db"]) < TOL


def test_pool_max_golden():
    g = load("pool_max")
    k, s, p = [int(v) for v in g["meta"]]
    y, mask = orc.pool_max_fwd(g["x"], k, k, p, p, s, s)
    assert y.shape == g["y"].shape
    assert relerr(y, g["y"]) < TOL
    dx = orc.pool_max_bwd(g["dy"], mask, g["x"].shape[2], g["x"].shape[3])
    assert relerr(dx, g["dx"]) < TOL


def test_pool_ave_golden():
    g = load("pool_ave")
    k, s, p = [int(v) for v in g["meta"]]
    y = orc.pool_ave_fwd(g["x"], k, k, p, p, s, s)
    assert relerr(y, g["y"]) < TOL
    dx = orc.pool_ave_bwd(g["dy"], g["x"].shape[2], g["x"].shape[3],
                          k, k, p, p, s, s)
    assert relerr(dx, g["dx"]) < TOL


def test_pool_reference_known_answer():
    # Restated as data from the reference's own TestForwardSquare
    # (src/caffe/test/test_pooling_layer.cpp:52-125): 2x2 channels of
    # [1 2 5 2 3; 9 4 1 4 8; 1 2 5 2 3], 2x2 max pool, stride 1.
    plane = np.array([[1, 2, 5, 2, 3],
                      [9, 4, 1, 4, 8],
                      [1, 2, 5, 2, 3]], np.float32)
    x = np.broadcast_to(plane, (2, 2, 3, 5)).copy()
    y, mask = orc.pool_max_fwd(x, 2, 2, 0, 0, 1, 1)
    want = np.array([[9, 5, 5, 8], [9, 5, 5, 8]], np.float32)
    assert (y == want).all()
    want_mask = np.array([[5, 2, 2, 9], [5, 12, 12, 9]], np.int32)
    assert (mask == want_mask).all()


def test_bn_golden():
    g = load("bn")
    eps = float(g["eps"][0])
    y, mean, var, inv_std, xnorm = orc.bn_fwd_train(g["x"], eps, g["scale"],
                                                    g["bias"])
    assert relerr(mean, g["mean"]) < TOL
    assert relerr(var, g["var"]) < TOL
    assert relerr(y, g["y"]) < TOL
    dx, dscale, dbias = orc.bn_bwd(xnorm, g["dy"], inv_std, g["scale"])
    assert relerr(dscale, g["dscale"]) < TOL
    assert relerr(dbias, g["dbias"]) < TOL
    assert relerr(dx, g["dx"]) < TOL


def test_softmaxloss_golden():
    g = load("softmaxloss")
    N, C = g["x"].shape
    prob = orc.softmax_fwd(g["x"], N, C, 1)
    assert relerr(prob, g["prob"]) < TOL
    loss = orc.softmaxloss_fwd(prob, g["label"], N, C, 1)
    assert abs(loss - g["loss"][0]) < 1e-5 * max(1, abs(g["loss"][0]))
    dx = orc.softmaxloss_bwd(prob, g["label"], N, C, 1)
    assert relerr(dx, g["dx"]) < TOL


def test_ip_golden():
    g = load("ip")
    y = orc.ip_fwd(g["x"], g["w"], g["b"])
    assert relerr(y, g["y"]) < TOL
    dx, dw, db = orc.ip_bwd(g["x"], g["w"], g["dy"])
    assert relerr(dx, g["dx"]) < TOL
    assert relerr(dw, g["dw"]) < TOL
    assert relerr(db, g["db"]) < TOL


def test_lrn_golden():
    g = load("lrn")
    size = int(g["meta"][0])
    alpha, beta, k = [float(v) for v in g["fmeta"]]
    y, scale = orc.lrn_fwd(g["x"], size, alpha, beta, k)
    assert relerr(y, g["y"]) < TOL
    dx = orc.lrn_bwd(g["x"], y, g["dy"], scale, size, alpha, beta)
    assert relerr(dx, g["dx"]) < TOL


def test_relu():
    x = np.random.default_rng(0).standard_normal(1000).astype(np.float32)
    dy = np.random.default_rng(1).standard_normal(1000).astype(np.float32)
    y = orc.relu_fwd(x)
    assert (y == np.maximum(x, 0)).all()
    dx = orc.relu_bwd(x, dy)
    assert (dx == dy * (x > 0)).all()


def test_im2col_col2im_roundtrip():
    rng = np.random.default_rng(3)
    im = rng.standard_normal((3, 9, 9)).astype(np.float32)
    col = orc.im2col(im, 3, 3, 1, 1, 2, 2)
    # col2im is the exact adjoint of im2col: <im2col(x), c> == <x, col2im(c)>
    c = rng.standard_normal(col.shape).astype(np.float32)
    back = orc.col2im(c, 3, 9, 9, 3, 3, 1, 1, 2, 2)
    lhs = float((col.astype(np.float64) * c).sum())
    rhs = float((im.astype(np.float64) * back).sum())
    assert abs(lhs - rhs) < 1e-3 * max(1.0, abs(lhs))


def test_sgd_update_closed_form():
    # Closed-form check in the spirit of test_gradient_based_solver.cpp:
    # g' = g*scale + decay*w ; h' = m*h + lr*g' ; w' = w - h' ; g'' = 0.
    rng = np.random.default_rng(5)
    n = 257
    g = rng.standard_normal(n).astype(np.float32)
    w = rng.standard_normal(n).astype(np.float32)
    h = rng.standard_normal(n).astype(np.float32)
    g0, w0, h0 = g.copy(), w.copy(), h.copy()
    m, lr, decay, scale = 0.9, 0.01, 5e-4, 0.125
    orc.sgd_update(g, w, h, m, lr, decay, scale)
    gp = g0 * scale + decay * w0
    hp = m * h0 + lr * gp
    assert relerr(h, hp) < 1e-6
    assert relerr(w, w0 - hp) < 1e-6
    assert (g == 0).all()


def test_accuracy():
    pred = np.array([[0.1, 0.9], [0.8, 0.2], [0.3, 0.7]], np.float32)
    lab = np.array([1, 1, 1], np.float32)
    assert abs(orc.accuracy(pred, lab, 3, 2, 1, 1) - 2.0 / 3) < 1e-6
    assert orc.accuracy(pred, lab, 3, 2, 1, 2) == 1.0


def test_golden_fixtures_bit_reproducible(tmp_path):
    # the committed .npz fixtures must regenerate bit-identically from
    # the committed generator (guards against silent fixture drift)
    import subprocess
    import sys
    env = dict(os.environ, GOLDEN_OUT=str(tmp_path))
    subprocess.check_call(
        [sys.executable, os.path.join(GOLD, "gen_golden.py")], env=env,
        stdout=subprocess.DEVNULL)
    regen = sorted(p.name for p in tmp_path.glob("*.npz"))
    committed = sorted(f for f in os.listdir(GOLD) if f.endswith(".npz"))
    assert regen == committed
    for name in regen:
        a = np.load(tmp_path / name)
        b = np.load(os.path.join(GOLD, name))
        assert set(a.files) == set(b.files), name
        for k in a.files:
            assert np.array_equal(a[k], b[k]), (name, k)
