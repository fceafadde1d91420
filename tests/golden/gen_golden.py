#!/usr/bin/env python3
"""Generate golden parity fixtures for the oracle using torch-CPU as the
secondary cross-check (the reference tree cannot be compiled
in this container — no protoc/boost/glog/gflags/MPI — so torch CPU pins the
oracle's arithmetic; reference-test known-answer constants are separately
restated as data in tests/test_oracle.py).

Run in the dev container (torch present): python3 tests/golden/gen_golden.py
Commits small .npz fixtures next to this script.  The GPU-box tests consume
the fixtures only — they never import this module.
"""
import os

import numpy as np
import torch
import torch.nn.functional as F

# output dir overridable so tests can verify the committed fixtures are
# bit-reproducible without touching them
HERE = os.environ.get("GOLDEN_OUT",
                      os.path.dirname(os.path.abspath(__file__)))
rng = np.random.default_rng(1371)


def t(a):
    return torch.from_numpy(a)


def save(name, **arrs):
    np.savez_compressed(os.path.join(HERE, name + ".npz"), **arrs)
    print(name, {k: v.shape for k, v in arrs.items()})


def gen_conv():
    cases = [
        # (N,C,H,W, Cout,k,s,p, group)  — shapes from the ResNet-50/AlexNet census
        (2, 3, 16, 16, 8, 7, 2, 3, 1),
        (2, 8, 14, 14, 16, 3, 1, 1, 1),
        (2, 16, 7, 7, 32, 1, 1, 0, 1),
        (2, 16, 9, 9, 32, 1, 2, 0, 1),
        (2, 8, 13, 13, 16, 5, 1, 2, 2),   # grouped (AlexNet conv2 class)
        (1, 4, 8, 8, 6, 3, 2, 0, 1),
    ]
    for i, (N, C, H, W, Co, k, s, p, g) in enumerate(cases):
        x = rng.standard_normal((N, C, H, W), np.float32)
        w = (rng.standard_normal((Co, C // g, k, k), np.float32) * 0.1)
        b = rng.standard_normal(Co, np.float32)
        tx = t(x).requires_grad_(True)
        tw = t(w).requires_grad_(True)
        tb = t(b).requires_grad_(True)
        y = F.conv2d(tx, tw, tb, stride=s, padding=p, groups=g)
        dy = rng.standard_normal(tuple(y.shape), np.float32)
        y.backward(t(dy))
        save(f"conv{i}",
             x=x, w=w, b=b, dy=dy, y=y.detach().numpy(),
             dx=tx.grad.numpy(), dw=tw.grad.numpy(), db=tb.grad.numpy(),
             meta=np.array([N, C, H, W, Co, k, s, p, g], np.int64))


def gen_pool():
    x = rng.standard_normal((2, 3, 13, 13), np.float32)
    tx = t(x).requires_grad_(True)
    y = F.max_pool2d(tx, 3, 2, 0, ceil_mode=True)
    dy = rng.standard_normal(tuple(y.shape), np.float32)
    y.backward(t(dy))
    save("pool_max", x=x, dy=dy, y=y.detach().numpy(), dx=tx.grad.numpy(),
         meta=np.array([3, 2, 0], np.int64))
    # global average pool (ResNet-50 head: 7x7)
    x2 = rng.standard_normal((2, 4, 7, 7), np.float32)
    tx2 = t(x2).requires_grad_(True)
    y2 = F.avg_pool2d(tx2, 7, 1, 0)
    dy2 = rng.standard_normal(tuple(y2.shape), np.float32)
    y2.backward(t(dy2))
    save("pool_ave", x=x2, dy=dy2, y=y2.detach().numpy(),
         dx=tx2.grad.numpy(), meta=np.array([7, 1, 0], np.int64))


def gen_bn():
    N, C, H, W = 4, 6, 5, 5
    eps = 1e-4  # note: layer clamps eps to >= 1e-5 (batch_norm_layer.cpp:25)
    x = rng.standard_normal((N, C, H, W), np.float32)
    sc = rng.standard_normal(C, np.float32) + 1.0
    bi = rng.standard_normal(C, np.float32)
    tx = t(x).requires_grad_(True)
    tsc = t(sc).requires_grad_(True)
    tbi = t(bi).requires_grad_(True)
    y = F.batch_norm(tx, torch.zeros(C), torch.ones(C), tsc, tbi,
                     training=True, momentum=0.0, eps=eps)
    dy = rng.standard_normal((N, C, H, W), np.float32)
    y.backward(t(dy))
    mean = x.mean(axis=(0, 2, 3))
    var = x.var(axis=(0, 2, 3), ddof=0)
    save("bn", x=x, scale=sc, bias=bi, dy=dy, y=y.detach().numpy(),
         dx=tx.grad.numpy(), dscale=tsc.grad.numpy(), dbias=tbi.grad.numpy(),
         mean=mean.astype(np.float32), var=var.astype(np.float32),
         eps=np.array([eps], np.float32))


def gen_softmaxloss():
    N, C = 8, 10
    x = rng.standard_normal((N, C), np.float32) * 3
    lab = rng.integers(0, C, N).astype(np.float32)
    tx = t(x).requires_grad_(True)
    loss = F.cross_entropy(tx, t(lab).long(), reduction="mean")
    loss.backward()
    prob = F.softmax(t(x), dim=1).numpy()
    save("softmaxloss", x=x, label=lab, prob=prob,
         loss=np.array([loss.item()], np.float32), dx=tx.grad.numpy())


def gen_ip():
    M, K, Nout = 5, 12, 7
    x = rng.standard_normal((M, K), np.float32)
    w = rng.standard_normal((Nout, K), np.float32)
    b = rng.standard_normal(Nout, np.float32)
    tx = t(x).requires_grad_(True)
    tw = t(w).requires_grad_(True)
    tb = t(b).requires_grad_(True)
    y = F.linear(tx, tw, tb)
    dy = rng.standard_normal((M, Nout), np.float32)
    y.backward(t(dy))
    save("ip", x=x, w=w, b=b, dy=dy, y=y.detach().numpy(),
         dx=tx.grad.numpy(), dw=tw.grad.numpy(), db=tb.grad.numpy())


def gen_lrn():
    # torch LocalResponseNorm matches caffe ACROSS_CHANNELS with alpha
    # pre-divided by size inside torch (torch divides by size too).
    N, C, H, W = 2, 8, 5, 5
    size, alpha, beta, k = 5, 1e-4, 0.75, 1.0
    x = rng.standard_normal((N, C, H, W), np.float32)
    tx = t(x).requires_grad_(True)
    y = F.local_response_norm(tx, size, alpha=alpha, beta=beta, k=k)
    dy = rng.standard_normal((N, C, H, W), np.float32)
    y.backward(t(dy))
    save("lrn", x=x, dy=dy, y=y.detach().numpy(), dx=tx.grad.numpy(),
         meta=np.array([size], np.int64),
         fmeta=np.array([alpha, beta, k], np.float32))


if __name__ == "__main__":
    gen_conv()
    gen_pool()
    gen_bn()
    gen_softmaxloss()
    gen_ip()
    gen_lrn()
