"""LMDB data pipeline: from-scratch reader (no liblmdb in
the environment) + Datum decode + rank-cycle sharding + crop/mirror/mean/
scale transform + prefetch worker.

The fixture database is written by tools/make_lmdb.py (pure python,
implementing the published LMDB page format); tests predict the record
bytes INDEPENDENTLY via the same splitmix64 recurrence and check the
engine's data blob end-to-end: mmap -> B+tree walk -> Datum wire decode
-> shard index -> transform.  Reference analogs: util/db_lmdb.cpp,
data_reader.cpp:295-301, data_transformer.cpp/.cu.
"""
import os
import struct
import subprocess
import sys

import numpy as np
import pytest

import caffe_amd as ca
from engine_util import REPO, net_from_text, relerr

sys.path.insert(0, os.path.join(REPO, "tools"))
from make_lmdb import make_lmdb, record_bytes  # noqa: E402

N_REC, C, H, W = 37, 3, 12, 12  # odd count exercises epoch wraparound
SEED = 99


@pytest.fixture(autouse=True)
def reset_stream():
    # the data-stream counter is engine-global and driven by solvers in
    # other tests — raw-net content checks need position 0 and rank 0/1
    ca.set_data_iter(0)
    ca.set_rank_world(0, 1)
    yield
    ca.set_data_iter(0)
    ca.set_rank_world(0, 1)


@pytest.fixture(scope="module")
def db(tmp_path_factory):
    d = tmp_path_factory.mktemp("lmdb") / "train_db"
    make_lmdb(str(d), N_REC, C, H, W, SEED)
    return str(d)


def data_net(db, batch=4, crop=0, mirror=False, scale=1.0, mean=None,
             phase="TRAIN"):
    tp = f"scale: {scale}\n"
    if crop:
        tp += f"    crop_size: {crop}\n"
    if mirror:
        tp += "    mirror: true\n"
    if mean is not None:
        tp += "".join(f"    mean_value: {m}\n" for m in mean)
    return f"""name: "t"
layer {{
  name: "data" type: "Data" top: "data" top: "label"
  include {{ phase: {phase} }}
  data_param {{ source: "{db}" batch_size: {batch} backend: LMDB }}
  transform_param {{
    {tp}  }}
}}
"""


def expected_image(idx):
    raw = np.frombuffer(record_bytes(SEED, idx, C, H, W),
                        np.uint8).astype(np.float32)
    return raw.reshape(C, H, W)


def test_reader_content_and_labels(db):
    # batch of 4 at iter 0, rank 0/world 1: records 0..3 verbatim
    ca.set_mode("cpu")
    net = net_from_text(data_net(db, batch=4, scale=0.5, mean=[10, 20, 30]))
    net.forward()
    data = np.asarray(net.blob("data")).reshape(4, C, H, W)
    labels = np.asarray(net.blob("label")).ravel()
    mean = np.array([10, 20, 30], np.float32).reshape(C, 1, 1)
    for j in range(4):
        exp = (expected_image(j) - mean) * 0.5
        assert relerr(data[j], exp) < 1e-6, j
        assert labels[j] == j % 10
    # shapes follow the datum, not the synthetic configuration
    assert list(net.blob_shape("data")) == [4, C, H, W]


def test_center_crop_test_phase(db):
    ca.set_mode("cpu")
    net = net_from_text(data_net(db, batch=2, crop=8, phase="TEST"),
                        phase=1)
    net.forward()
    data = np.asarray(net.blob("data")).reshape(2, C, 8, 8)
    off = (H - 8) // 2
    for j in range(2):
        exp = expected_image(j)[:, off:off + 8, off:off + 8]
        assert relerr(data[j], exp) < 1e-6


def test_train_crop_in_bounds_and_deterministic(db):
    ca.set_mode("cpu")
    outs = []
    for _ in range(2):
        ca.set_random_seed(123)
        net = net_from_text(data_net(db, batch=6, crop=8, mirror=True))
        net.forward()
        outs.append(np.asarray(net.blob("data")).copy())
    assert np.array_equal(outs[0], outs[1])  # keyed RNG: reproducible
    # every row must be an 8x8 window of the right record (possibly
    # mirrored) — check membership against all candidate crops
    data = outs[0].reshape(6, C, 8, 8)
    for j in range(2):
        img = expected_image(j)
        cands = []
        for ho in range(H - 8 + 1):
            for wo in range(W - 8 + 1):
                win = img[:, ho:ho + 8, wo:wo + 8]
                cands.append(win)
                cands.append(win[:, :, ::-1])
        assert any(np.array_equal(data[j], c) for c in cands), j


def test_rank_sharding_disjoint(db):
    # world 2: rank r reads records (iter*batch+j)*2 + r — disjoint and
    # together covering the even/odd interleave (data_reader.cpp:295-301
    # ownership semantics)
    ca.set_mode("cpu")
    seen = {}
    for rank in (0, 1):
        ca.set_rank_world(rank, 2)
        net = net_from_text(data_net(db, batch=4))
        net.forward()
        d = np.asarray(net.blob("data")).reshape(4, C, H, W)
        seen[rank] = d
        for j in range(4):
            assert relerr(d[j], expected_image((j * 2 + rank) % N_REC)) \
                < 1e-6
    ca.set_rank_world(0, 1)
    assert not np.array_equal(seen[0], seen[1])


def test_epoch_wrap_and_solver_resume(db, tmp_path):
    # records cycle mod N_REC (37): a solver stepping past one epoch keeps
    # training, and snapshot/restore resumes the exact record stream
    ca.set_mode("cpu")
    ca.set_rank_world(0, 1)
    text = f"""base_lr: 0.01
lr_policy: "fixed"
momentum: 0.9
random_seed: 5
snapshot_prefix: "{tmp_path}/s"
net_param {{
  name: "n"
{data_net(db, batch=8, crop=8).split(chr(10), 1)[1]}
  layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
    inner_product_param {{ num_output: 10
      weight_filler {{ type: "gaussian" std: 0.05 }} }} }}
  layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
    bottom: "label" top: "loss" }}
}}
"""
    s = ca.Solver(text=text)
    s.step(8)  # 64 samples > 37 records: wraps
    assert np.isfinite(s.loss())
    assert ca._lib.caffe_solver_snapshot(s._h) == 0
    state = os.path.join(tmp_path, "s_iter_8.solverstate")
    s.step(2)
    ref = [s.net.param(i).copy() for i in range(s.net.num_params())]
    s2 = ca.Solver(text=text)
    assert ca._lib.caffe_solver_restore(s2._h, state.encode()) == 0
    s2.step(2)
    for i, r in enumerate(ref):
        assert relerr(s2.net.param(i), r) < 1e-6, i


def test_overflow_values():
    # records large enough to spill into LMDB overflow pages
    import tempfile
    d = tempfile.mkdtemp() + "/big_db"
    make_lmdb(d, 5, 3, 40, 40, 7)  # 4800B datum > half a page
    ca.set_mode("cpu")
    ca.set_rank_world(0, 1)
    net = net_from_text(data_net(d, batch=3))
    net.forward()
    data = np.asarray(net.blob("data")).reshape(3, 3, 40, 40)
    for j in range(3):
        raw = np.frombuffer(record_bytes(7, j, 3, 40, 40),
                            np.uint8).astype(np.float32).reshape(3, 40, 40)
        assert relerr(data[j], raw) < 1e-6


def test_dist_lmdb_shard_equivalence(db):
    # world-2 LMDB-sharded training (gloo callback comm) == 1-rank on the
    # doubled batch: the per-iteration record SET is identical (rank r owns
    # (iter*B+j)*2+r), so averaged gradients match — the reference's
    # effective-batch equivalence extended to the LMDB feed.  Full-image
    # crops keep the rank-keyed crop RNG out of the comparison.
    import subprocess
    import sys as _sys
    from test_dist_cpu import parse_params, run_dist

    worker = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "_dist_lmdb_worker.py")
    out2 = run_dist(2, [db, "4", "6"], worker=worker)
    out1 = subprocess.run(
        [_sys.executable, worker, db, "8", "6"],
        env=dict(os.environ, RANK="0", WORLD_SIZE="1"), cwd=REPO,
        capture_output=True, text=True, timeout=600)
    assert out1.returncode == 0, out1.stdout + out1.stderr
    p2 = parse_params(out2[0])
    p1 = parse_params(out1.stdout)
    assert np.abs(p2 - p1).max() < 1e-4 * max(1.0, np.abs(p1).max())


@pytest.mark.gpu
def test_lmdb_gpu_transform_parity(db):
    # GPU path (pinned upload + k_transform_u8) against the CPU transform
    ca.set_rank_world(0, 1)
    outs = {}
    for mode in ("cpu", "gpu"):
        ca.set_mode(mode)
        ca.set_random_seed(31)
        net = net_from_text(data_net(db, batch=6, crop=8, mirror=True,
                                     scale=0.0078125, mean=[104, 117, 123]))
        net.forward()
        outs[mode] = np.asarray(net.blob("data")).copy()
        labs = np.asarray(net.blob("label")).ravel()
        assert labs.tolist() == [j % 10 for j in range(6)]
    assert relerr(outs["gpu"], outs["cpu"]) < 1e-6


@pytest.mark.gpu
def test_lmdb_gpu_training(db):
    # short GPU training run off the LMDB feed (prefetch worker + device
    # transform in the real solver loop)
    ca.set_mode("gpu")
    ca.set_rank_world(0, 1)
    text = f"""base_lr: 0.01
lr_policy: "fixed"
momentum: 0.9
random_seed: 5
net_param {{
  name: "n"
{data_net(db, batch=8, crop=8).split(chr(10), 1)[1]}
  layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
    inner_product_param {{ num_output: 10
      weight_filler {{ type: "gaussian" std: 0.05 }} }} }}
  layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
    bottom: "label" top: "loss" }}
}}
"""
    s = ca.Solver(text=text)
    s.step(10)
    assert np.isfinite(s.loss())


def test_cli_caffe_test_on_lmdb(db, tmp_path):
    # the `caffe test` subcommand scoring a TEST-phase LMDB net (CPU):
    # reference tools/caffe.cpp:241 test() — forward test_iter batches,
    # print averaged scores
    import subprocess
    model = tmp_path / "net.prototxt"
    model.write_text(f"""name: "t"
layer {{
  name: "data" type: "Data" top: "data" top: "label"
  include {{ phase: TEST }}
  data_param {{ source: "{db}" batch_size: 5 backend: LMDB }}
  transform_param {{ scale: 0.02 }}
}}
layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
  inner_product_param {{ num_output: 10
    weight_filler {{ type: "gaussian" std: 0.1 }} }} }}
layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "label"
  top: "loss" }}
layer {{ name: "acc" type: "Accuracy" bottom: "fc" bottom: "label"
  top: "accuracy" include {{ phase: TEST }} }}
""")
    caffe_bin = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")
    r = subprocess.run([caffe_bin, "test", f"-model={model}",
                        "-iterations=3"], capture_output=True, text=True,
                       timeout=300, cwd=REPO)
    assert r.returncode == 0, r.stdout + r.stderr
    out = r.stdout + r.stderr
    assert "loss" in out and "acc" in out, out
    import re
    acc = float(re.search(r"acc = ([\d.]+)", out).group(1))
    assert 0.0 <= acc <= 1.0


def test_cli_test_walks_distinct_batches(db, tmp_path):
    # regression: TestAll/`caffe test` must advance the data cursor per
    # iteration — before the fix every test iteration scored ONE batch
    import re
    import subprocess
    model = tmp_path / "tnet.prototxt"
    model.write_text(f"""name: "t"
layer {{
  name: "data" type: "Data" top: "data" top: "label"
  include {{ phase: TEST }}
  data_param {{ source: "{db}" batch_size: 5 backend: LMDB }}
  transform_param {{ scale: 0.02 }}
}}
layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
  inner_product_param {{ num_output: 10
    weight_filler {{ type: "gaussian" std: 0.3 }} }} }}
layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "label"
  top: "loss" }}
""")
    caffe_bin = os.path.join(REPO, "caffe-mpi.github.io_amd", "caffe")

    def cli_loss(iters):
        r = subprocess.run(
            [caffe_bin, "test", f"-model={model}", f"-iterations={iters}"],
            capture_output=True, text=True, timeout=300, cwd=REPO,
            env=dict(os.environ, CAFFE_SEED="1371"))
        assert r.returncode == 0, r.stdout + r.stderr
        return float(re.search(r"loss = ([\d.]+)",
                               r.stdout + r.stderr).group(1))

    l1 = cli_loss(1)   # records 0..4
    l3 = cli_loss(3)   # records 0..14 averaged
    # independent expectation: forward the same TEST net at cursor 0,1,2
    ca.set_mode("cpu")
    ca.set_random_seed(1371)
    net = net_from_text(model.read_text(), phase=1)
    losses = []
    for it in range(3):
        ca.set_data_iter(it)
        net.forward()
        losses.append(float(np.asarray(net.blob("loss")).ravel()[0]))
    # weights differ between CLI and this net (different fill order), so
    # only the STRUCTURAL property is pinned: 3-iteration average uses 3
    # distinct batches (it must differ from the single-batch loss, and
    # our own 3-batch losses must not all be equal)
    assert len({round(v, 6) for v in losses}) > 1, losses
    assert abs(l3 - l1) > 1e-6, (l1, l3)


def test_mean_file_binaryproto(db, tmp_path):
    # transform_param.mean_file: a BlobProto (binaryproto) full-size mean,
    # subtracted in SOURCE coordinates — encoded here independently
    # (field 7 shape{dim}, field 5 packed float data)
    from make_lmdb import varint
    rng = np.random.default_rng(77)
    mean = rng.uniform(0, 255, (C, H, W)).astype(np.float32)
    shape_msg = b""
    for d in (C, H, W):
        shape_msg += b"\x08" + varint(d)
    blob = b"\x3a" + varint(len(shape_msg)) + shape_msg  # field 7 (shape)
    data = mean.ravel().tobytes()
    blob += b"\x2a" + varint(len(data)) + data           # field 5 packed
    mf = tmp_path / "mean.binaryproto"
    mf.write_bytes(blob)
    ca.set_mode("cpu")
    net = net_from_text(f"""name: "t"
layer {{
  name: "data" type: "Data" top: "data" top: "label"
  data_param {{ source: "{db}" batch_size: 2 backend: LMDB }}
  transform_param {{ mean_file: "{mf}" scale: 0.5 }}
}}
""")
    net.forward()
    out = np.asarray(net.blob("data")).reshape(2, C, H, W)
    for j in range(2):
        exp = (expected_image(j) - mean) * 0.5
        assert relerr(out[j], exp) < 1e-5, j


def test_iter_size_equivalence_on_lmdb(db, tmp_path):
    # iter_size 2 @ batch 4 == batch 8: the sub-passes advance the data
    # cursor (data_iter = iter*iter_size + sub), so both runs consume the
    # SAME record set per update — gradients and params must match
    def train(iter_size, batch, tag):
        ca.set_mode("cpu")
        ca.set_rank_world(0, 1)
        ca.set_data_iter(0)
        text = f"""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 6
iter_size: {iter_size}
snapshot_prefix: "{tmp_path}/{tag}"
net_param {{
  name: "n"
  layer {{ name: "data" type: "Data" top: "data" top: "label"
    data_param {{ source: "{db}" batch_size: {batch} backend: LMDB }}
    transform_param {{ scale: 0.0078125 mean_value: 128 }} }}
  layer {{ name: "ip" type: "InnerProduct" bottom: "data" top: "fc"
    inner_product_param {{ num_output: 10
      weight_filler {{ type: "gaussian" std: 0.05 }} }} }}
  layer {{ name: "loss" type: "SoftmaxWithLoss" bottom: "fc"
    bottom: "label" top: "loss" }}
}}
"""
        s = ca.Solver(text=text)
        s.step(4)
        return [np.asarray(s.net.param(i)).copy()
                for i in range(s.net.num_params())]

    acc = train(2, 4, "a")
    ref = train(1, 8, "b")
    for a, b in zip(acc, ref):
        assert relerr(a, b) < 1e-5


def test_corrupt_lmdb_errors(tmp_path):
    # reader failures at setup must be loud and name the problem:
    # (a) wrong magic, (b) file shorter than the two meta pages
    bad = tmp_path / "bad_lmdb"
    bad.mkdir()
    (bad / "data.mdb").write_bytes(b"\x00" * 8192)
    net_text = f"""name: "t"
layer {{ name: "data" type: "Data" top: "data" top: "label"
  data_param {{ source: "{bad}" batch_size: 2 }} }}
"""
    ca.set_mode("cpu")
    with pytest.raises(Exception) as e:
        net_from_text(net_text)
    assert "not an LMDB file" in str(e.value)

    short = tmp_path / "short_lmdb"
    short.mkdir()
    (short / "data.mdb").write_bytes(b"\x01\x02")
    net_text2 = net_text.replace(str(bad), str(short))
    with pytest.raises(Exception):
        net_from_text(net_text2)
