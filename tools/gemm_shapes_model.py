import sys, os
sys.path.insert(0, "caffe-mpi.github.io_amd")
os.environ["CAFFE_GEMM_BY_SHAPE"] = "1"
import caffe_amd as ca
model = sys.argv[1] if len(sys.argv) > 1 else "googlenet"
batch = int(sys.argv[2]) if len(sys.argv) > 2 else 128
ca.set_mode("gpu", 0)
ca.set_synthetic_shape(3, 227 if model == "alexnet" else 224, 227 if model == "alexnet" else 224, 1000)
ca.set_random_seed(1371)
s = ca.Solver(path=f"models/generated/{model}_solver.prototxt", batch_override=batch)
s.step(3)
ca.device_synchronize()
ca.perf_reset()
ca.set_perf_timing(True)
s.step(3)
ca.device_synchronize()
perf = ca.perf_snapshot()
rows = [(k, v["ns"]/3e6, v["flops"]/max(v["ns"],1)/1e3, v["launches"])
        for k, v in perf.items() if v["ns"] > 0]
rows.sort(key=lambda r: -r[1])
tot = sum(r[1] for r in rows)
print(f"total {tot:.1f} ms/step")
for k, ms, tf, n in rows[:24]:
    print(f"{k:30s} {ms:7.2f} ms {tf:6.1f} TF {n:4d}")
