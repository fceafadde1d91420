"""pycaffe-compatible API shim (reference python/caffe: _caffe.cpp +
pycaffe.py).  A script written against classic pycaffe —

    import caffe
    caffe.set_mode_gpu(); caffe.set_device(0)
    net = caffe.Net(proto, caffe.TEST)           # or (proto, weights, TEST)
    net.blobs['data'].data[...] = x
    net.forward(); net.backward()
    w = net.params['conv1'][0].data
    solver = caffe.SGDSolver(solver_proto)
    solver.step(100); solver.net.blobs['loss'].data

— runs against this engine by `import caffe_amd.pycaffe as caffe`.

Blob.data / Blob.diff are ZERO-COPY numpy views over the engine's host
mirror with the reference's SyncedMemory semantics: reading `.data`
syncs device->host; every access through the `data` property re-marks
the host copy authoritative exactly like pycaffe's mutable_cpu_data
binding, so `net.blobs['x'].data[...] = v` is visible to the next GPU
forward.  Only the training hot-path surface is implemented (no
layer-by-layer `forward(start=,end=)` slicing, no HDF5) — the reference
boost::python module itself is OUT of hot-path scope.
"""
import ctypes
from collections import OrderedDict

import numpy as np

from . import CaffeError, Net as _Net, Solver as _Solver, _ck, _lib

TRAIN = 0
TEST = 1

_lib.caffe_net_blob_cpu_ptr.restype = ctypes.POINTER(ctypes.c_float)
_lib.caffe_net_blob_cpu_ptr.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                        ctypes.c_int, ctypes.c_int]
_lib.caffe_net_num_layers.argtypes = [ctypes.c_void_p]
_lib.caffe_net_blob_names.argtypes = [ctypes.c_void_p, ctypes.c_char_p,
                                      ctypes.c_int]
_lib.caffe_net_layer_info.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                      ctypes.c_char_p, ctypes.c_int,
                                      ctypes.c_char_p, ctypes.c_int,
                                      ctypes.POINTER(ctypes.c_int)]
_lib.caffe_net_layer_blob_shape.argtypes = [
    ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int,
    ctypes.POINTER(ctypes.c_int), ctypes.c_int,
    ctypes.POINTER(ctypes.c_int)]
_lib.caffe_net_layer_blob_cpu_ptr.restype = ctypes.POINTER(ctypes.c_float)
_lib.caffe_net_layer_blob_cpu_ptr.argtypes = [
    ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int, ctypes.c_int,
    ctypes.c_int]


def set_mode_cpu():
    _ck(_lib.caffe_set_mode(0, 0))


def set_mode_gpu():
    _ck(_lib.caffe_set_mode(1, _device[0]))


_device = [0]


def set_device(d):
    _device[0] = d
    _ck(_lib.caffe_set_mode(1, d))


def set_random_seed(s):
    _ck(_lib.caffe_set_random_seed(s))


class _BlobView:
    """pycaffe Blob: .data/.diff as shaped numpy views, .shape, .count."""

    def __init__(self, fetch_ptr, fetch_shape):
        self._ptr = fetch_ptr      # (diff, writable) -> float*
        self._shape = fetch_shape  # () -> tuple

    @property
    def shape(self):
        return self._shape()

    @property
    def count(self):
        n = 1
        for d in self._shape():
            n *= d
        return int(n)

    def _arr(self, diff):
        p = self._ptr(diff, 1)
        if not p:
            raise CaffeError(_lib.caffe_last_error().decode())
        shape = self._shape()
        n = self.count
        a = np.ctypeslib.as_array(p, shape=(max(n, 1),))
        return a[:n].reshape(shape if shape else (1,))

    @property
    def data(self):
        return self._arr(0)

    @property
    def diff(self):
        return self._arr(1)


class _LayerView:
    def __init__(self, type_):
        self.type = type_


class Net:
    """caffe.Net(proto, phase) or caffe.Net(proto, weights, phase)."""

    def __init__(self, proto, *args):
        if len(args) == 1:
            weights, phase = None, args[0]
        elif len(args) == 2:
            weights, phase = args
        else:
            raise TypeError("Net(proto, [weights,] phase)")
        self._net = _Net.from_file(proto, phase=phase)
        if weights:
            self._net.load_weights(weights)
        self._refresh()

    def _refresh(self):
        h = self._net._h
        buf = ctypes.create_string_buffer(65536)
        _ck(_lib.caffe_net_blob_names(h, buf, 65536))
        names = [s for s in buf.value.decode().split("\n") if s]
        self.blobs = OrderedDict()
        for name in names:
            self.blobs[name] = _BlobView(
                lambda diff, w, nm=name: _lib.caffe_net_blob_cpu_ptr(
                    h, nm.encode(), diff, w),
                lambda nm=name: tuple(self._net.blob_shape(nm)))
        self.params = OrderedDict()
        self.layers = []
        self._layer_names = []
        nbuf = ctypes.create_string_buffer(256)
        tbuf = ctypes.create_string_buffer(256)
        nb = ctypes.c_int()
        for i in range(_lib.caffe_net_num_layers(h)):
            _ck(_lib.caffe_net_layer_info(h, i, nbuf, 256, tbuf, 256,
                                          ctypes.byref(nb)))
            lname = nbuf.value.decode()
            self._layer_names.append(lname)
            self.layers.append(_LayerView(tbuf.value.decode()))
            if nb.value > 0:
                self.params[lname] = [
                    _BlobView(
                        lambda diff, w, ln=lname, bi=j:
                            _lib.caffe_net_layer_blob_cpu_ptr(
                                h, ln.encode(), bi, diff, w),
                        lambda ln=lname, bi=j: self._layer_blob_shape(
                            ln, bi))
                    for j in range(nb.value)]

    def _layer_blob_shape(self, lname, bidx):
        shape = (ctypes.c_int * 8)()
        nd = ctypes.c_int()
        _ck(_lib.caffe_net_layer_blob_shape(self._net._h, lname.encode(),
                                            bidx, shape, 8,
                                            ctypes.byref(nd)))
        return tuple(shape[i] for i in range(nd.value))

    # pycaffe returns a dict of output blobs from forward()
    def forward(self):
        self._net.forward()
        return {}

    def backward(self):
        self._net.backward()
        return {}

    def save(self, path):
        self._net.save_weights(path)

    def copy_from(self, path):
        self._net.load_weights(path)

    @property
    def _blob_names(self):
        return list(self.blobs.keys())


class SGDSolver:
    """caffe.SGDSolver(prototxt) — Solver::Step semantics."""

    def __init__(self, path):
        self._solver_path = path
        self._solver = _Solver(path=path)
        self.net = _WrappedSolverNet(self._solver)

    def step(self, iters):
        self._solver.step(iters)

    def solve(self):
        # reference Solver::Solve: run to SolverParameter.max_iter
        import re
        mi = 0
        try:
            txt = open(self._solver_path).read()
            m = re.search(r"max_iter:\s*(\d+)", txt)
            mi = int(m.group(1)) if m else 0
        except Exception:
            pass
        if mi:
            self._solver.step(mi)

    def snapshot(self):
        _ck(_lib.caffe_solver_snapshot(self._solver._h))

    def restore(self, state):
        _ck(_lib.caffe_solver_restore(self._solver._h, state.encode()))

    @property
    def iter(self):
        return self._solver.iter


class _WrappedSolverNet(Net):
    """the solver's train net, wrapped without re-creating it."""

    def __init__(self, solver):
        self._net = solver.net
        self._refresh()


# module-level alias so `import caffe_amd.pycaffe as caffe` reads like
# classic `import caffe`
Classifier = None  # out of scope (deploy-time helper, needs image IO)
