import sys, os, json
sys.path.insert(0, "caffe-mpi.github.io_amd")
os.environ["CAFFE_GEMM_BY_SHAPE"] = "1"
import caffe_amd as ca
ca.set_mode("gpu", 0)
ca.set_synthetic_shape(3, 224, 224, 1000)
ca.set_random_seed(1371)
s = ca.Solver(path="models/generated/resnet50_solver.prototxt", batch_override=128)
s.step(3)
ca.device_synchronize()
ca.perf_reset()
ca.set_perf_timing(True)
s.step(3)
ca.device_synchronize()
perf = ca.perf_snapshot()
rows = [(k, v["ns"]/3e6, v["flops"]/max(v["ns"],1)/1e3, v["launches"])
        for k, v in perf.items() if k.startswith("gemm")]
rows.sort(key=lambda r: -r[1])
for k, ms, tf, n in rows[:30]:
    print(f"{k:28s} {ms:7.2f} ms/step {tf:6.1f} TF {n:4d}")
