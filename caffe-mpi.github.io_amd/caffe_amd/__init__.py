"""caffe_amd — host-side mirror of the Caffe-MPI Layer/Net/Solver surface.

ctypes binding over libcaffe_amd.so's C ABI (include/caffe_amd.h; each entry
point there cites the reference interface it replaces).  PyTorch is used by
callers only for torch.distributed rendezvous — the engine itself is C++/HIP.
"""
import ctypes
import os
import subprocess

import numpy as np

_PKG = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO = os.path.join(_PKG, "libcaffe_amd.so")


def _load():
    if not os.path.exists(_SO):
        subprocess.check_call(["make", "-C", _PKG, "-j8"])
    try:
        return ctypes.CDLL(_SO)
    except OSError as e:
        raise ImportError(
            f"libcaffe_amd.so failed to load ({e}); the HIP engine is "
            "mandatory — no fallback") from e


_lib = _load()

_lib.caffe_last_error.restype = ctypes.c_char_p
# declare handle argtypes (void*) — without these ctypes truncates to int
_vp = ctypes.c_void_p
_lib.caffe_solver_create.argtypes = [ctypes.c_char_p, ctypes.c_int]
_lib.caffe_solver_create_from_text.argtypes = [ctypes.c_char_p, ctypes.c_int]
_lib.caffe_solver_free.argtypes = [_vp]
_lib.caffe_solver_step.argtypes = [_vp, ctypes.c_int]
_lib.caffe_solver_iter.argtypes = [_vp]
_lib.caffe_solver_loss.argtypes = [_vp]
_lib.caffe_solver_net.argtypes = [_vp]
_lib.caffe_comm_unique_id.argtypes = [ctypes.POINTER(ctypes.c_uint8)]
_lib.caffe_comm_init.argtypes = [_vp, ctypes.c_int, ctypes.c_int,
                                 ctypes.POINTER(ctypes.c_uint8)]
_lib.caffe_comm_bcast_weights.argtypes = [_vp]
_lib.caffe_net_create.argtypes = [ctypes.c_char_p, ctypes.c_int,
                                  ctypes.c_int]
_lib.caffe_net_free.argtypes = [_vp]
_lib.caffe_net_forward.argtypes = [_vp]
_lib.caffe_net_backward.argtypes = [_vp]
_lib.caffe_net_loss.argtypes = [_vp]
_lib.caffe_net_blob_shape.argtypes = [_vp, ctypes.c_char_p,
                                      ctypes.POINTER(ctypes.c_int),
                                      ctypes.c_int,
                                      ctypes.POINTER(ctypes.c_int)]
_lib.caffe_net_num_params.argtypes = [_vp]
_lib.caffe_solver_snapshot.argtypes = [_vp]
_lib.caffe_solver_restore.argtypes = [_vp, ctypes.c_char_p]
_lib.caffe_net_save_weights.argtypes = [_vp, ctypes.c_char_p]
_lib.caffe_net_load_weights.argtypes = [_vp, ctypes.c_char_p]
_lib.caffe_solver_create.restype = ctypes.c_void_p
_lib.caffe_solver_create_from_text.restype = ctypes.c_void_p
_lib.caffe_solver_net.restype = ctypes.c_void_p
_lib.caffe_solver_loss.restype = ctypes.c_float
_lib.caffe_solver_iter.restype = ctypes.c_long
_lib.caffe_net_create.restype = ctypes.c_void_p
_lib.caffe_net_loss.restype = ctypes.c_float
_lib.caffe_set_random_seed.argtypes = [ctypes.c_uint64]
_lib.caffe_set_compute.argtypes = [ctypes.c_char_p]

_f32p = ctypes.POINTER(ctypes.c_float)
_lib.caffe_net_blob_get.argtypes = [_vp, ctypes.c_char_p, ctypes.c_int,
                                    _f32p, ctypes.c_long]
_lib.caffe_net_blob_set.argtypes = [_vp, ctypes.c_char_p, ctypes.c_int,
                                    _f32p, ctypes.c_long]
_lib.caffe_net_param_info.argtypes = [_vp, ctypes.c_int, ctypes.c_char_p,
                                      ctypes.c_int,
                                      ctypes.POINTER(ctypes.c_int),
                                      ctypes.POINTER(ctypes.c_long)]
_lib.caffe_net_param_get.argtypes = [_vp, ctypes.c_int, ctypes.c_int, _f32p,
                                     ctypes.c_long]
_lib.caffe_net_param_set.argtypes = [_vp, ctypes.c_int, _f32p, ctypes.c_long]

ALLREDUCE_CB = ctypes.CFUNCTYPE(None, _f32p, ctypes.c_long, ctypes.c_void_p)
_lib.caffe_comm_set_callback.argtypes = [_vp, ALLREDUCE_CB, ctypes.c_void_p,
                                         ctypes.c_int]


class CaffeError(RuntimeError):
    pass


def _ck(rc):
    """status-code return: 0 = ok"""
    if rc is None or rc != 0:
        raise CaffeError(_lib.caffe_last_error().decode())
    return rc


def _ckp(ptr):
    """pointer return: NULL = error"""
    if not ptr:
        raise CaffeError(_lib.caffe_last_error().decode())
    return ptr


def set_mode(mode, device=0):
    _ck(_lib.caffe_set_mode(1 if mode == "gpu" else 0, device))


def set_data_iter(i):
    """Data-stream position (LMDB cursor / synthetic counter)."""
    _ck(_lib.caffe_set_data_iter(ctypes.c_uint64(i)))


def set_rank_world(rank, world):
    """Engine rank/world for the sharded data feed (no communicator)."""
    _ck(_lib.caffe_set_rank_world(rank, world))


def set_compute(dtype):
    """GEMM compute dtype: "f32" (exact, default) or "bf16" (bf16 MFMA
    with fp32 accumulation — mixed precision; storage stays fp32)."""
    _ck(_lib.caffe_set_compute(dtype.encode()))


def set_random_seed(seed):
    _ck(_lib.caffe_set_random_seed(seed))


def set_synthetic_shape(c, h, w, num_classes=1000):
    _ck(_lib.caffe_set_synthetic_shape(c, h, w, num_classes))


def set_perf_timing(enable):
    _ck(_lib.caffe_set_perf_timing(1 if enable else 0))


def device_synchronize():
    _ck(_lib.caffe_device_synchronize())


def perf_snapshot():
    cap, rows = 64, 256
    names = ctypes.create_string_buffer(cap * rows)
    launches = (ctypes.c_long * rows)()
    flops = (ctypes.c_double * rows)()
    bytes_ = (ctypes.c_double * rows)()
    ns = (ctypes.c_double * rows)()
    n = _lib.caffe_perf_snapshot(names, cap, launches, flops, bytes_, ns,
                                 rows)
    if n < 0:
        raise CaffeError(_lib.caffe_last_error().decode())
    out = {}
    for i in range(n):
        name = names.raw[i * cap:(i + 1) * cap].split(b"\0")[0].decode()
        out[name] = dict(launches=launches[i], flops=flops[i],
                         bytes=bytes_[i], ns=ns[i])
    return out


def perf_reset():
    _ck(_lib.caffe_perf_reset())


class Net:
    def __init__(self, handle, owned=False):
        self._h = handle
        self._owned = owned

    @classmethod
    def from_file(cls, path, phase=0, batch_override=0):
        h = _ckp(_lib.caffe_net_create(path.encode(), phase, batch_override))
        return cls(h, owned=True)

    def forward(self):
        _ck(_lib.caffe_net_forward(self._h))

    def backward(self):
        _ck(_lib.caffe_net_backward(self._h))

    def loss(self):
        v = _lib.caffe_net_loss(self._h)
        if v == -1.0:
            err = _lib.caffe_last_error().decode()
            if err:
                raise CaffeError(err)
        return v

    def blob_shape(self, name):
        shape = (ctypes.c_int * 8)()
        nd = ctypes.c_int()
        _ck(_lib.caffe_net_blob_shape(self._h, name.encode(), shape, 8,
                                      ctypes.byref(nd)))
        return tuple(shape[i] for i in range(nd.value))

    def blob(self, name, diff=False):
        shape = self.blob_shape(name)
        out = np.empty(shape if shape else (1,), np.float32)
        _ck(_lib.caffe_net_blob_get(
            self._h, name.encode(), int(diff),
            out.ctypes.data_as(_f32p), out.size))
        return out

    def set_blob(self, name, arr, diff=False):
        arr = np.ascontiguousarray(arr, np.float32)
        _ck(_lib.caffe_net_blob_set(
            self._h, name.encode(), int(diff),
            arr.ctypes.data_as(_f32p), arr.size))

    def num_params(self):
        return _lib.caffe_net_num_params(self._h)

    def param_info(self, idx):
        buf = ctypes.create_string_buffer(256)
        bi = ctypes.c_int()
        cnt = ctypes.c_long()
        _ck(_lib.caffe_net_param_info(self._h, idx, buf, 256,
                                      ctypes.byref(bi), ctypes.byref(cnt)))
        return buf.value.decode(), bi.value, cnt.value

    def param(self, idx, diff=False):
        _, _, cnt = self.param_info(idx)
        out = np.empty(cnt, np.float32)
        _ck(_lib.caffe_net_param_get(self._h, idx, int(diff),
                                     out.ctypes.data_as(_f32p), cnt))
        return out

    def set_param(self, idx, arr):
        arr = np.ascontiguousarray(arr, np.float32)
        _ck(_lib.caffe_net_param_set(self._h, idx,
                                     arr.ctypes.data_as(_f32p), arr.size))

    def params(self):
        return {i: self.param_info(i) for i in range(self.num_params())}

    def save_weights(self, path):
        """NetParameter binaryproto with every layer blob (incl. BN stats)
        — net.cpp ToProto / tools/caffe.cpp -weights surface."""
        _ck(_lib.caffe_net_save_weights(self._h, path.encode()))

    def load_weights(self, path):
        _ck(_lib.caffe_net_load_weights(self._h, path.encode()))


class Solver:
    def __init__(self, path=None, text=None, batch_override=0):
        if path is not None:
            self._h = _ckp(_lib.caffe_solver_create(path.encode(),
                                                   batch_override))
        else:
            self._h = _ckp(_lib.caffe_solver_create_from_text(
                text.encode(), batch_override))
        self.net = Net(_ckp(_lib.caffe_solver_net(self._h)))
        self._cb_keepalive = None

    def step(self, iters=1):
        _ck(_lib.caffe_solver_step(self._h, iters))

    @property
    def iter(self):
        return _lib.caffe_solver_iter(self._h)

    def loss(self):
        return _lib.caffe_solver_loss(self._h)

    def comm_unique_id(self):
        buf = (ctypes.c_uint8 * 128)()
        _ck(_lib.caffe_comm_unique_id(buf))
        return bytes(buf)

    def comm_init(self, rank, world, uid_bytes):
        buf = (ctypes.c_uint8 * 128)(*uid_bytes)
        _ck(_lib.caffe_comm_init(self._h, rank, world, buf))

    def bcast_weights(self):
        _ck(_lib.caffe_comm_bcast_weights(self._h))

    def set_allreduce_callback(self, fn, world):
        """CPU-mode collective for gloo tests: fn(np_array) reduced in place."""
        def _trampoline(ptr, count, _ud):
            arr = np.ctypeslib.as_array(ptr, shape=(count,))
            fn(arr)
        cb = ALLREDUCE_CB(_trampoline)
        self._cb_keepalive = cb
        _ck(_lib.caffe_comm_set_callback(self._h, cb, None, world))
