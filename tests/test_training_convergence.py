"""End-to-end TRAINING signal, beyond numerics parity: a small fixed
LMDB dataset must be memorized (loss collapses) by a conv net under the
real solver loop — forward, backward, bucketed reducer, fused SGD — in
fp32 and in bf16 mixed precision.  A kernel bug that preserves layerwise
parity on one step but corrupts state across steps (momentum, moving
averages, diff arena reuse) shows up here.
"""
import os
import sys

import numpy as np
import pytest

import caffe_amd as ca
from engine_util import REPO

sys.path.insert(0, os.path.join(REPO, "tools"))
from make_lmdb import make_lmdb  # noqa: E402


def solver_text(db, tmp):
    return f"""base_lr: 0.02
lr_policy: "fixed"
momentum: 0.9
random_seed: 4
snapshot_prefix: "{tmp}/s"
net_param {{
  name: "n"
  layer {{ name: "data" type: "Data" top: "data" top: "label"
    data_param {{ source: "{db}" batch_size: 16 backend: LMDB }}
    transform_param {{ scale: 0.0078125 mean_value: 128 }} }}
  layer {{ name: "c1" type: "Convolution" bottom: "data" top: "c1"
    convolution_param {{ num_output: 8 kernel_size: 3 pad: 1
      weight_filler {{ type: "msra" }} }} }}
  layer {{ name: "bn1" type: "BatchNorm" bottom: "c1" top: "c1"
    batch_norm_param {{ scale_bias: true }} }}
  layer {{ name: "r1" type: "ReLU" bottom: "c1" top: "c1" }}
  layer {{ name: "p1" type: "Pooling" bottom: "c1" top: "p1"
    pooling_param {{ pool: MAX kernel_size: 2 stride: 2 }} }}
  layer {{ name: "ip" type: "InnerProduct" bottom: "p1" top: "fc"
    inner_product_param {{ num_output: 10
      weight_filler {{ type: "xavier" }} }} }}
  layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
    bottom: "label" top: "loss" }}
}}
"""


def run_training(mode, dtype, db, tmp, iters=120):
    ca.set_mode(mode)
    if mode == "gpu":
        ca.set_compute(dtype)
    ca.set_rank_world(0, 1)
    ca.set_data_iter(0)
    s = ca.Solver(text=solver_text(db, tmp))
    s.step(10)
    early = s.loss()
    s.step(iters - 10)
    late = s.loss()
    if mode == "gpu":
        ca.set_compute("f32")
    return early, late


@pytest.fixture(scope="module")
def db16(tmp_path_factory):
    # 16 records = exactly one batch: pure memorization target
    d = tmp_path_factory.mktemp("conv_db") / "db"
    make_lmdb(str(d), 16, 3, 16, 16, 321)
    return str(d)


def test_cpu_overfit(db16, tmp_path):
    early, late = run_training("cpu", "f32", db16, tmp_path, iters=60)
    assert np.isfinite(late)
    assert late < 0.5 * early, (early, late)


@pytest.mark.gpu
def test_gpu_overfit_f32(db16, tmp_path):
    early, late = run_training("gpu", "f32", db16, tmp_path)
    assert np.isfinite(late)
    assert late < 0.3 * early, (early, late)


@pytest.mark.gpu
def test_gpu_overfit_bf16(db16, tmp_path):
    early, late = run_training("gpu", "bf16", db16, tmp_path)
    assert np.isfinite(late)
    assert late < 0.3 * early, (early, late)
