"""Multi-node RCCL bootstrap (csrc/bootstrap.cpp): the 128-byte
ncclUniqueId exchange over TCP — the reference's Clusters/MPI_Bcast
replacement (clusters.cpp:8, parallel.cpp:42-45).  No GPUs or multiple
boxes needed to pin the exchange itself: a serve thread plays global
rank 0, fetchers play the other ranks of both nodes.
"""
import ctypes
import socket
import threading

import numpy as np

import caffe_amd as ca

_lib = ca._lib
_lib.caffe_uid_serve.argtypes = [ctypes.POINTER(ctypes.c_uint8),
                                 ctypes.c_int, ctypes.c_int]
_lib.caffe_uid_fetch.argtypes = [ctypes.POINTER(ctypes.c_uint8),
                                 ctypes.c_char_p, ctypes.c_int,
                                 ctypes.c_int]


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_uid_exchange_three_clients():
    rng = np.random.default_rng(8)
    uid = (ctypes.c_uint8 * 128)(*rng.integers(0, 256, 128,
                                               dtype=np.uint8))
    port = _free_port()
    rc_serve = []

    def serve():
        rc_serve.append(_lib.caffe_uid_serve(uid, port, 3))

    t = threading.Thread(target=serve)
    t.start()
    got = []
    for _ in range(3):  # the "other ranks", possibly on other nodes
        out = (ctypes.c_uint8 * 128)()
        rc = _lib.caffe_uid_fetch(out, b"127.0.0.1", port, 10)
        assert rc == 0
        got.append(bytes(out))
    t.join(timeout=30)
    assert rc_serve == [0]
    for g in got:
        assert g == bytes(uid)


def test_fetch_retries_until_server_up():
    # fetcher starts BEFORE the server (the usual multi-node race):
    # the retry loop must win once rank 0 binds
    uid = (ctypes.c_uint8 * 128)(*range(100, 228))
    port = _free_port()
    out = (ctypes.c_uint8 * 128)()
    res = []

    def fetch():
        res.append(_lib.caffe_uid_fetch(out, b"127.0.0.1", port, 20))

    t = threading.Thread(target=fetch)
    t.start()
    import time
    time.sleep(0.5)
    assert _lib.caffe_uid_serve(uid, port, 1) == 0
    t.join(timeout=30)
    assert res == [0]
    assert bytes(out) == bytes(uid)


def test_fetch_timeout():
    out = (ctypes.c_uint8 * 128)()
    assert _lib.caffe_uid_fetch(out, b"127.0.0.1", _free_port(), 1) != 0
