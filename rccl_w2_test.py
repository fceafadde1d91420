"""World-2 RCCL on ONE device (both ranks device 0): if RCCL accepts it,
run the real bucketed-reducer training equivalence on GPU."""
import os, sys, subprocess
REPO = "/root/repo"
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))

def worker(rank):
    import numpy as np
    import caffe_amd as ca
    ca.set_mode("gpu", 0)
    ca.set_rank_world(rank, 2)
    ca.set_random_seed(9)
    text = """base_lr: 0.05
lr_policy: "fixed"
momentum: 0.9
random_seed: 9
net_param {
  name: "n"
  layer { name: "input" type: "Input" top: "in0" top: "in1"
    input_param { shape { dim: 4 dim: 3 dim: 8 dim: 8 } shape { dim: 4 } } }
  layer { name: "c1" type: "Convolution" bottom: "in0" top: "c"
    convolution_param { num_output: 6 kernel_size: 3
      weight_filler { type: "gaussian" std: 0.2 } } }
  layer { name: "ip" type: "InnerProduct" bottom: "c" top: "fc"
    inner_product_param { num_output: 5
      weight_filler { type: "gaussian" std: 0.2 } } }
  layer { name: "loss" type: "SoftmaxWithLoss" bottom: "fc" bottom: "in1"
    top: "loss" }
}
"""
    s = ca.Solver(text=text)
    # uid via file
    import time
    uidp = "/tmp/w2uid"
    if rank == 0:
        uid = s.comm_unique_id()
        open(uidp + ".tmp", "wb").write(uid)
        os.rename(uidp + ".tmp", uidp)
    else:
        for _ in range(600):
            if os.path.exists(uidp): break
            time.sleep(0.1)
        uid = open(uidp, "rb").read()
    s.comm_init(rank, 2, uid)
    s.bcast_weights()
    import numpy as np
    rng = np.random.default_rng(77)
    for i in range(net_iters):
        x = rng.standard_normal((4,3,8,8)).astype(np.float32)
        l = rng.integers(0,5,4).astype(np.float32)
        # SAME data on both ranks -> allreduce avg == single-rank grads
        s.net.set_blob("in0", x); s.net.set_blob("in1", l)
        s.step(1)
    vals = np.concatenate([np.asarray(s.net.param(i)).ravel() for i in range(s.net.num_params())])
    print(f"RANK{rank} PARAMS", " ".join(f"{v:.6e}" for v in vals[::11][:20]))

net_iters = 4
if __name__ == "__main__":
    if len(sys.argv) > 1:
        worker(int(sys.argv[1]))
    else:
        try: os.remove("/tmp/w2uid")
        except OSError: pass
        ps = [subprocess.Popen([sys.executable, __file__, str(r)],
              stdout=subprocess.PIPE, stderr=subprocess.STDOUT) for r in (0,1)]
        outs = [p.communicate(timeout=300)[0].decode() for p in ps]
        for p, o in zip(ps, outs):
            print("rc", p.returncode)
            print(o[-500:])
