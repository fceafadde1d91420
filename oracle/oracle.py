"""ctypes wrapper over liboracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this module (build contract).  See oracle/oracle.c for the reference
citations each function restates.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def _load():
    if not os.path.exists(_SO):
        subprocess.check_call(["make", "-C", _DIR])
    return ctypes.CDLL(_SO)


_lib = _load()

_f32p = ctypes.POINTER(ctypes.c_float)
_i32p = ctypes.POINTER(ctypes.c_int)
_u32p = ctypes.POINTER(ctypes.c_uint)


def _fp(a):
    assert a.dtype == np.float32 and a.flags["C_CONTIGUOUS"]
    return a.ctypes.data_as(_f32p)


def _ip(a):
    assert a.dtype == np.int32 and a.flags["C_CONTIGUOUS"]
    return a.ctypes.data_as(_i32p)


_lib.orc_conv_out_dim.restype = ctypes.c_int
_lib.orc_softmaxloss_fwd.restype = ctypes.c_float
_lib.orc_accuracy.restype = ctypes.c_float


def conv_out_dim(i, k, p, s, d=1):
    return _lib.orc_conv_out_dim(i, k, p, s, d)


def im2col(im, kh, kw, ph, pw, sh, sw, dh=1, dw=1):
    C, H, W = im.shape
    OH, OW = conv_out_dim(H, kh, ph, sh, dh), conv_out_dim(W, kw, pw, sw, dw)
    col = np.empty((C * kh * kw, OH * OW), np.float32)
    _lib.orc_im2col(_fp(im), C, H, W, kh, kw, ph, pw, sh, sw, dh, dw, _fp(col))
    return col


def col2im(col, C, H, W, kh, kw, ph, pw, sh, sw, dh=1, dw=1):
    im = np.empty((C, H, W), np.float32)
    _lib.orc_col2im(_fp(col), C, H, W, kh, kw, ph, pw, sh, sw, dh, dw, _fp(im))
    return im


def gemm(A, B, transA=False, transB=False, alpha=1.0, beta=0.0, C=None):
    M = A.shape[1] if transA else A.shape[0]
    K = A.shape[0] if transA else A.shape[1]
    N = B.shape[0] if transB else B.shape[1]
    if C is None:
        C = np.zeros((M, N), np.float32)
    _lib.orc_gemm(int(transA), int(transB), M, N, K, ctypes.c_float(alpha),
                  _fp(A), _fp(B), ctypes.c_float(beta), _fp(C))
    return C


def conv_fwd(x, w, b=None, pad=(0, 0), stride=(1, 1), dil=(1, 1), group=1):
    N, C, H, W = x.shape
    Cout, _, kh, kw = w.shape
    OH = conv_out_dim(H, kh, pad[0], stride[0], dil[0])
    OW = conv_out_dim(W, kw, pad[1], stride[1], dil[1])
    y = np.empty((N, Cout, OH, OW), np.float32)
    colbuf = np.empty((C * kh * kw, OH * OW), np.float32)
    wb = np.ascontiguousarray(w.reshape(Cout, -1))
    _lib.orc_conv_fwd(_fp(x), _fp(wb), _fp(b) if b is not None else _f32p(),
                      int(b is not None), N, C, H, W, Cout, kh, kw,
                      pad[0], pad[1], stride[0], stride[1], dil[0], dil[1],
                      group, _fp(y), _fp(colbuf))
    return y


def conv_bwd(x, w, dy, pad=(0, 0), stride=(1, 1), dil=(1, 1), group=1,
             want_dx=True, want_dw=True, want_db=False):
    N, C, H, W = x.shape
    Cout, _, kh, kw = w.shape
    _, _, OH, OW = dy.shape
    dx = np.zeros_like(x)
    wb = np.ascontiguousarray(w.reshape(Cout, -1))
    dwb = np.zeros_like(wb)
    db = np.zeros(Cout, np.float32)
    colbuf = np.empty((C * kh * kw, OH * OW), np.float32)
    colbuf2 = np.empty_like(colbuf)
    _lib.orc_conv_bwd(_fp(x), _fp(wb), _fp(dy), N, C, H, W, Cout, kh, kw,
                      pad[0], pad[1], stride[0], stride[1], dil[0], dil[1],
                      group, int(want_dx), int(want_dw), int(want_db),
                      _fp(dx), _fp(dwb), _fp(db), _fp(colbuf), _fp(colbuf2))
    return dx, dwb.reshape(w.shape), db


def pool_out_dim(H, W, kh, kw, ph, pw, sh, sw):
    oh = ctypes.c_int()
    ow = ctypes.c_int()
    _lib.orc_pool_out_dim(H, W, kh, kw, ph, pw, sh, sw,
                          ctypes.byref(oh), ctypes.byref(ow))
    return oh.value, ow.value


def pool_max_fwd(x, kh, kw, ph, pw, sh, sw):
    N, C, H, W = x.shape
    OH, OW = pool_out_dim(H, W, kh, kw, ph, pw, sh, sw)
    y = np.empty((N, C, OH, OW), np.float32)
    mask = np.empty((N, C, OH, OW), np.int32)
    _lib.orc_pool_max_fwd(_fp(x), N, C, H, W, kh, kw, ph, pw, sh, sw,
                          _fp(y), _ip(mask))
    return y, mask


def pool_max_bwd(dy, mask, H, W):
    N, C, OH, OW = dy.shape
    dx = np.empty((N, C, H, W), np.float32)
    _lib.orc_pool_max_bwd(_fp(dy), _ip(mask), N, C, H, W, OH, OW, _fp(dx))
    return dx


def pool_ave_fwd(x, kh, kw, ph, pw, sh, sw):
    N, C, H, W = x.shape
    OH, OW = pool_out_dim(H, W, kh, kw, ph, pw, sh, sw)
    y = np.empty((N, C, OH, OW), np.float32)
    _lib.orc_pool_ave_fwd(_fp(x), N, C, H, W, kh, kw, ph, pw, sh, sw, _fp(y))
    return y


def pool_ave_bwd(dy, H, W, kh, kw, ph, pw, sh, sw):
    N, C = dy.shape[:2]
    dx = np.empty((N, C, H, W), np.float32)
    _lib.orc_pool_ave_bwd(_fp(dy), N, C, H, W, kh, kw, ph, pw, sh, sw,
                          _fp(dx))
    return dx


def bn_fwd_train(x, eps, scale=None, bias=None):
    N, C = x.shape[:2]
    S = int(np.prod(x.shape[2:])) if x.ndim > 2 else 1
    sb = scale is not None
    mean = np.empty(C, np.float32)
    var = np.empty(C, np.float32)
    inv_std = np.empty(C, np.float32)
    xnorm = np.empty_like(x)
    y = np.empty_like(x)
    _lib.orc_bn_fwd_train(_fp(x), N, C, S, ctypes.c_float(eps),
                          _fp(scale) if sb else _f32p(),
                          _fp(bias) if sb else _f32p(), int(sb),
                          _fp(mean), _fp(var), _fp(inv_std), _fp(xnorm),
                          _fp(y))
    return y, mean, var, inv_std, xnorm


def bn_fwd_test(x, eps, gmean, gvar, scale=None, bias=None):
    N, C = x.shape[:2]
    S = int(np.prod(x.shape[2:])) if x.ndim > 2 else 1
    sb = scale is not None
    y = np.empty_like(x)
    _lib.orc_bn_fwd_test(_fp(x), N, C, S, ctypes.c_float(eps), _fp(gmean),
                         _fp(gvar), _fp(scale) if sb else _f32p(),
                         _fp(bias) if sb else _f32p(), int(sb), _fp(y))
    return y


def bn_bwd(xnorm, dy, inv_std, scale=None):
    N, C = xnorm.shape[:2]
    S = int(np.prod(xnorm.shape[2:])) if xnorm.ndim > 2 else 1
    sb = scale is not None
    dx = np.empty_like(xnorm)
    dscale = np.zeros(C, np.float32)
    dbias = np.zeros(C, np.float32)
    _lib.orc_bn_bwd(_fp(xnorm), _fp(dy), _fp(inv_std),
                    _fp(scale) if sb else _f32p(), int(sb), N, C, S,
                    _fp(dx), _fp(dscale), _fp(dbias))
    return dx, dscale, dbias


def relu_fwd(x, slope=0.0):
    y = np.empty_like(x)
    _lib.orc_relu_fwd(_fp(x), ctypes.c_long(x.size), ctypes.c_float(slope),
                      _fp(y))
    return y


def relu_bwd(x, dy, slope=0.0):
    dx = np.empty_like(x)
    _lib.orc_relu_bwd(_fp(x), _fp(dy), ctypes.c_long(x.size),
                      ctypes.c_float(slope), _fp(dx))
    return dx


def lrn_fwd(x, size, alpha, beta, k=1.0):
    N, C, H, W = x.shape
    scale = np.empty_like(x)
    y = np.empty_like(x)
    _lib.orc_lrn_fwd(_fp(x), N, C, H, W, size, ctypes.c_float(alpha),
                     ctypes.c_float(beta), ctypes.c_float(k), _fp(scale),
                     _fp(y))
    return y, scale


def lrn_bwd(x, y, dy, scale, size, alpha, beta):
    N, C, H, W = x.shape
    dx = np.empty_like(x)
    _lib.orc_lrn_bwd(_fp(x), _fp(y), _fp(dy), _fp(scale), N, C, H, W, size,
                     ctypes.c_float(alpha), ctypes.c_float(beta), _fp(dx))
    return dx


def softmax_fwd(x, outer, C, inner):
    y = np.empty_like(x)
    _lib.orc_softmax_fwd(_fp(x), outer, C, inner, _fp(y))
    return y


def softmaxloss_fwd(prob, label, outer, C, inner):
    return _lib.orc_softmaxloss_fwd(_fp(prob), _fp(label), outer, C, inner,
                                    0, 0)


def softmaxloss_bwd(prob, label, outer, C, inner, loss_weight=1.0):
    dx = np.empty_like(prob)
    _lib.orc_softmaxloss_bwd(_fp(prob), _fp(label), outer, C, inner, 0, 0,
                             ctypes.c_float(loss_weight), _fp(dx))
    return dx


def ip_fwd(x, w, b=None):
    M, K = x.shape
    Nout = w.shape[0]
    y = np.empty((M, Nout), np.float32)
    _lib.orc_ip_fwd(_fp(x), _fp(w), _fp(b) if b is not None else _f32p(),
                    int(b is not None), M, Nout, K, _fp(y))
    return y


def ip_bwd(x, w, dy, want_dx=True, has_bias=True):
    M, K = x.shape
    Nout = w.shape[0]
    dx = np.zeros_like(x)
    dw = np.zeros_like(w)
    db = np.zeros(Nout, np.float32)
    _lib.orc_ip_bwd(_fp(x), _fp(w), _fp(dy), M, Nout, K, int(want_dx),
                    _fp(dx), _fp(dw), _fp(db), int(has_bias))
    return dx, dw, db


def sgd_update(g, w, h, momentum, lr, decay, grad_scale=1.0):
    _lib.orc_sgd_update(ctypes.c_long(g.size), _fp(g), _fp(w), _fp(h),
                        ctypes.c_float(momentum), ctypes.c_float(lr),
                        ctypes.c_float(decay), ctypes.c_float(grad_scale))


def accuracy(pred, label, outer, C, inner, top_k=1):
    return _lib.orc_accuracy(_fp(pred), _fp(label), outer, C, inner, top_k)
