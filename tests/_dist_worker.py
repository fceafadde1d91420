"""Worker for test_dist_cpu.py — one process per rank, CPU engine, gloo
collective plugged into the engine's bucket-flush callback."""
import argparse
import os
import sys
import tempfile

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))

import caffe_amd as ca  # noqa: E402

SOLVER_TEXT = """
snapshot_prefix: "SNAPPFX"
base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
weight_decay: 0.0005
random_seed: 5
net_param {
  name: "distnet"
  layer {
    name: "input"
    type: "Input"
    top: "in0"
    top: "in1"
    input_param {
      shape { dim: %d dim: 3 dim: 8 dim: 8 }
      shape { dim: %d }
    }
  }
  layer {
    name: "c1"
    type: "Convolution"
    bottom: "in0"
    top: "c1"
    convolution_param { num_output: 4 kernel_size: 3 pad: 1 }
  }
  layer { name: "r1" type: "ReLU" bottom: "c1" top: "c1" }
  layer {
    name: "ip"
    type: "InnerProduct"
    bottom: "c1"
    top: "fc"
    inner_product_param { num_output: 5 }
  }
  layer {
    name: "loss"
    type: "SoftmaxWithLoss"
    bottom: "fc"
    bottom: "in1"
    top: "loss"
  }
}
"""


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--rank-data", choices=["same", "shard", "combined"],
                    default="same")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--iter-size", type=int, default=1)
    ap.add_argument("--restore-at", type=int, default=-1,
                    help="snapshot at this iter, rebuild the solver, "
                         "restore, re-attach the comm, continue")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))

    ca.set_mode("cpu")
    tmp = tempfile.mkdtemp()
    text = (SOLVER_TEXT % (args.batch, args.batch)).replace(
        "SNAPPFX", os.path.join(tmp, f"r{rank}"))
    if args.iter_size > 1:
        text = f"iter_size: {args.iter_size}\n" + text
    solver = ca.Solver(text=text)
    net = solver.net

    if world > 1:
        import torch
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)

        def reduce_fn(arr):
            t = torch.from_numpy(arr)
            dist.all_reduce(t)  # sums in place

        solver.set_allreduce_callback(reduce_fn, world)

    # identical init on every rank (the reference bcasts rank-0 weights,
    # parallel.cpp:208-227; callback comm has no bcast so we pin directly)
    prng = np.random.default_rng(77)
    for i in range(net.num_params()):
        _, _, cnt = net.param_info(i)
        net.set_param(i, (prng.standard_normal(cnt) * 0.1).astype(np.float32))

    pool_b = 16
    for it in range(args.iters):
        if it == args.restore_at:
            # mid-training restart: snapshot, tear the solver down, build a
            # fresh one, Restore (params + momentum history + iter), and
            # re-attach the collective — the 8-GPU resume path
            # (solver.cpp:542 Snapshot / :604 Restore semantics)
            assert ca._lib.caffe_solver_snapshot(solver._h) == 0
            state = os.path.join(tmp, f"r{rank}_iter_{solver.iter}"
                                      ".solverstate")
            assert os.path.exists(state), state
            solver = ca.Solver(text=text)
            net = solver.net
            assert ca._lib.caffe_solver_restore(solver._h,
                                                state.encode()) == 0
            if world > 1:
                solver.set_allreduce_callback(reduce_fn, world)
        rng = np.random.default_rng(1000 + it)
        data = rng.standard_normal((pool_b, 3, 8, 8)).astype(np.float32)
        labels = rng.integers(0, 5, pool_b).astype(np.float32)
        if args.rank_data == "same":
            d, l = data[:args.batch], labels[:args.batch]
        elif args.rank_data == "shard":
            d = data[rank * args.batch:(rank + 1) * args.batch]
            l = labels[rank * args.batch:(rank + 1) * args.batch]
        else:  # combined
            d, l = data[:args.batch], labels[:args.batch]
        net.set_blob("in0", d)
        net.set_blob("in1", l)
        solver.step(1)

    if rank == 0:
        vals = []
        for i in range(net.num_params()):
            vals.extend(net.param(i)[:8].tolist())
        print("PARAMS " + " ".join(f"{v:.8e}" for v in vals))
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
