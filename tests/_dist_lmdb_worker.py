"""Worker for the world-2 LMDB-sharded training test (gloo callback comm,
CPU).  Each rank owns the record stream (iter*batch+j)*world + rank; with
full-image crops (no crop RNG) a 2-rank run must equal a 1-rank run on the
doubled batch — the records per iteration are the SAME SET, so the
averaged gradients match (the reference's effective-batch equivalence
extended to the LMDB feed)."""
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))

import caffe_amd as ca  # noqa: E402


def main():
    db = sys.argv[1]
    batch = int(sys.argv[2])
    iters = int(sys.argv[3])
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))

    ca.set_mode("cpu")
    ca.set_rank_world(rank, world)
    text = f"""base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 11
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
    solver = ca.Solver(text=text)
    if world > 1:
        import torch
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)

        def reduce_fn(arr):
            t = torch.from_numpy(arr)
            dist.all_reduce(t)

        solver.set_allreduce_callback(reduce_fn, world)

    # identical init on every rank (callback comm has no bcast)
    prng = np.random.default_rng(77)
    net = solver.net
    for i in range(net.num_params()):
        _, _, cnt = net.param_info(i)
        net.set_param(i, (prng.standard_normal(cnt) * 0.1)
                      .astype(np.float32))
    solver.step(iters)
    if rank == 0:
        vals = np.concatenate([np.asarray(net.param(i)).ravel()
                               for i in range(net.num_params())])
        print("PARAMS " + " ".join(f"{v:.8e}" for v in vals[::7][:64]))


if __name__ == "__main__":
    main()
