"""bench.py's own multi-rank plumbing, exercised at world 2 on CPU
(CAFFE_BENCH_CPU=1): gloo rendezvous, warmup/timed barriers,
max-over-ranks reduction and the single rank-0 JSON line.  The RCCL leg
is covered separately (world-1 GPU selftest + the driver's 8-GPU run);
this pins everything else in the script the scaling bench depends on.
"""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def run_bench_world2():
    env = dict(os.environ, CAFFE_BENCH_CPU="1",
               MASTER_ADDR="127.0.0.1", MASTER_PORT=str(_free_port()))
    procs = []
    for rank in range(2):
        env_r = dict(env, RANK=str(rank), WORLD_SIZE="2",
                     LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "bench.py"),
             "--gpus", "2", "--steps", "1", "--warmup", "0",
             "--batch", "2"],
            env=env_r, cwd=REPO,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=900)
        assert p.returncode == 0, err.decode()[-2000:]
        outs.append(out.decode())
    return outs


def test_bench_world2_cpu_dryrun():
    outs = run_bench_world2()
    # exactly one JSON line, from rank 0
    json_lines = [ln for o in outs for ln in o.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, outs
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["metric"] == "images/sec"
    assert d["roofline"]["peak"] > 0
