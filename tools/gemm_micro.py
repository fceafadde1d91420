import sys, os, time
sys.path.insert(0, "caffe-mpi.github.io_amd")
sys.path.insert(0, "tests")
import numpy as np
import caffe_amd as ca
from engine_util import net_from_text, input_net

ca.set_mode("gpu", 0)
M = K = Nout = 4096
body = f"""layer {{
  name: "ip"
  type: "InnerProduct"
  bottom: "in0"
  top: "out"
  inner_product_param {{ num_output: {Nout} bias_term: false }}
}}"""
net = net_from_text(input_net([(M, K)], body))
rng = np.random.default_rng(0)
net.set_blob("in0", rng.standard_normal((M, K)).astype(np.float32))
net.set_param(0, (rng.standard_normal(Nout * K) * 0.05).astype(np.float32))
net.forward()  # warmup + upload
net.set_blob("out", rng.standard_normal((M, Nout)).astype(np.float32), diff=True)
net.backward()
ca.device_synchronize()
ca.perf_reset()
ca.set_perf_timing(True)
for _ in range(5):
    net.forward()   # NT
    net.backward()  # TN (dW) + NN (dx)
ca.device_synchronize()
perf = ca.perf_snapshot()
for k, v in perf.items():
    if k.startswith("gemm") and v["ns"] > 0:
        print(k, "TF:", round(v["flops"] / v["ns"] / 1e3, 1),
              "ms:", round(v["ns"] / 1e6, 2), "launches:", v["launches"])
