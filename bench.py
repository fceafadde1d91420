#!/usr/bin/env python3
"""bench.py — headline benchmark: ResNet-50 fp32 training images/sec
(BASELINE.json configs[2] at N=1: 1×MI355X, synthetic 3×224×224, per-GPU
batch 128; N>1 = configs[3], weak scaling, RCCL all-reduce over xGMI).

Contract: python bench.py --gpus N --steps K --warmup W
(N>1 ranks launched by torch.distributed.run; gloo is used only for
rendezvous/ncclUniqueId exchange and the timing barrier — the engine's own
RCCL communicator does the gradient collectives.)
Prints ONE JSON line from rank 0.
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.join(REPO, "caffe-mpi.github.io_amd"))
sys.path.insert(0, REPO)

PEAK_F32_MFMA = 157.3e12  # gfx950 fp32 matrix peak (spec; 155 TF measured)
# bf16 dense MFMA peak: 32x32x16 bf16 = 1024 FLOP/clk/SIMD x 4 x 256 CU
# x 2.4 GHz (MI355X_MICROARCH; AMD's 5 PF headline is 2:1-sparse)
PEAK_BF16_MFMA = 2516.6e12


def oracle_resnet50_step(batch):
    """CPU baseline leg: ONE fwd+bwd+update of ResNet-50 through oracle/
    (the reference CPU_ONLY semantics restated — kind 'port').  Returns
    seconds for the step.  Bounded: a small batch keeps this ~10-30 s."""
    import numpy as np
    from oracle import oracle as orc

    spec = json.load(open(os.path.join(REPO, "models", "specs",
                                       "resnet50.json")))
    rng = np.random.default_rng(1371)

    def dget(msg, key, dflt=None):
        for k, v in msg:
            if k == key:
                return v
        return dflt

    layers = []
    for k, v in spec:
        if k != "layer":
            continue
        d = {kk: vv for kk, vv in v}
        inc = dget(v, "include")
        if inc and dget(inc, "phase") == "TEST":
            continue
        if d.get("type") in ("Accuracy",):
            continue
        layers.append((d, v))

    blobs = {}
    saved = []  # per-layer context for backward
    x = rng.standard_normal((batch, 3, 224, 224)).astype(np.float32)
    labels = rng.integers(0, 1000, batch).astype(np.float32)
    t0 = time.time()
    for d, v in layers:
        t = d["type"]
        if t == "Data":
            blobs["data"] = x
            blobs["label"] = labels
            saved.append(None)
            continue
        bots = [b for k, b in v if k == "bottom"]
        tops = [b for k, b in v if k == "top"]
        if t == "Convolution":
            cp = {k: v2 for k, v2 in dget(v, "convolution_param")}
            kk = int(cp.get("kernel_size", 1))
            s = int(cp.get("stride", 1))
            p = int(cp.get("pad", 0))
            co = int(cp["num_output"])
            bx = blobs[bots[0]]
            w = (rng.standard_normal(
                (co, bx.shape[1], kk, kk)) *
                 np.sqrt(2.0 / (bx.shape[1] * kk * kk))).astype(np.float32)
            y = orc.conv_fwd(bx, w, None, pad=(p, p), stride=(s, s))
            blobs[tops[0]] = y
            saved.append(("conv", bots[0], tops[0], bx, w, (p, s)))
        elif t == "BatchNorm":
            bx = blobs[bots[0]]
            C = bx.shape[1]
            sc = np.ones(C, np.float32)
            bi = np.zeros(C, np.float32)
            y, mean, var, inv_std, xnorm = orc.bn_fwd_train(bx, 1e-4, sc, bi)
            blobs[tops[0]] = y
            saved.append(("bn", bots[0], tops[0], xnorm, inv_std, sc))
        elif t == "ReLU":
            bx = blobs[bots[0]]
            y = orc.relu_fwd(bx)
            blobs[tops[0]] = y
            saved.append(("relu", bots[0], tops[0], bx))
        elif t == "Pooling":
            pp = {k: v2 for k, v2 in dget(v, "pooling_param")}
            bx = blobs[bots[0]]
            if pp.get("global_pooling") == "true":
                kk = bx.shape[2]
            else:
                kk = int(pp.get("kernel_size", 1))
            s = int(pp.get("stride", 1))
            if pp.get("pool", "MAX") == "MAX":
                y, mask = orc.pool_max_fwd(bx, kk, kk, 0, 0, s, s)
                saved.append(("poolmax", bots[0], tops[0], mask,
                              bx.shape[2:], (kk, s)))
            else:
                y = orc.pool_ave_fwd(bx, kk, kk, 0, 0, s, s)
                saved.append(("poolave", bots[0], tops[0], bx.shape[2:],
                              (kk, s)))
            blobs[tops[0]] = y
        elif t == "Eltwise":
            y = blobs[bots[0]] + blobs[bots[1]]
            blobs[tops[0]] = y
            saved.append(("sum", bots, tops[0]))
        elif t == "InnerProduct":
            ip = {k: v2 for k, v2 in dget(v, "inner_product_param")}
            n = int(ip["num_output"])
            bx = blobs[bots[0]].reshape(batch, -1)
            w = (rng.standard_normal((n, bx.shape[1])) * 0.01).astype(
                np.float32)
            b = np.zeros(n, np.float32)
            y = orc.ip_fwd(bx, w, b)
            blobs[tops[0]] = y
            saved.append(("ip", bots[0], tops[0], bx, w,
                          blobs[bots[0]].shape))
        elif t == "SoftmaxWithLoss":
            bx = blobs[bots[0]]
            prob = orc.softmax_fwd(bx, batch, bx.shape[1], 1)
            loss = orc.softmaxloss_fwd(prob, blobs[bots[1]], batch,
                                       bx.shape[1], 1)
            saved.append(("loss", bots[0], prob, loss))
        else:
            raise RuntimeError(f"oracle walker: unhandled layer {t}")

    # backward
    diffs = {}

    def add_diff(name, d):
        if name in diffs:
            diffs[name] = diffs[name] + d
        else:
            diffs[name] = d

    for entry in reversed(saved):
        if entry is None:
            continue
        kind = entry[0]
        if kind == "loss":
            _, bot, prob, _ = entry
            add_diff(bot, orc.softmaxloss_bwd(prob, labels, batch,
                                              prob.shape[1], 1))
        elif kind == "ip":
            _, bot, top, bx, w, shape = entry
            dy = diffs[top]
            dx, dw, db = orc.ip_bwd(bx, w, dy)
            orc.sgd_update(dw.ravel(), w.ravel(), np.zeros_like(w.ravel()),
                           0.9, 0.001, 1e-4)
            add_diff(bot, dx.reshape(shape))
        elif kind == "sum":
            _, bots, top = entry
            dy = diffs[top]
            add_diff(bots[0], dy)
            add_diff(bots[1], dy)
        elif kind == "poolmax":
            _, bot, top, mask, hw, (kk, s) = entry
            add_diff(bot, orc.pool_max_bwd(diffs[top], mask, hw[0], hw[1]))
        elif kind == "poolave":
            _, bot, top, hw, (kk, s) = entry
            add_diff(bot, orc.pool_ave_bwd(diffs[top], hw[0], hw[1], kk, kk,
                                           0, 0, s, s))
        elif kind == "relu":
            _, bot, top, bx = entry
            dy = diffs[top]
            if top == bot:  # in-place ReLU: replace, don't accumulate
                diffs[bot] = orc.relu_bwd(bx, dy)
            else:
                add_diff(bot, orc.relu_bwd(bx, dy))
        elif kind == "bn":
            _, bot, top, xnorm, inv_std, sc = entry
            dx, dsc, dbi = orc.bn_bwd(xnorm, diffs[top], inv_std, sc)
            add_diff(bot, dx)
        elif kind == "conv":
            _, bot, top, bx, w, (p, s) = entry
            dy = diffs[top]
            want_dx = bot != "data"
            dx, dw, db = orc.conv_bwd(bx, w, dy, pad=(p, p), stride=(s, s),
                                      want_dx=want_dx)
            orc.sgd_update(dw.ravel(), w.ravel(), np.zeros_like(w.ravel()),
                           0.9, 0.001, 1e-4)
            if want_dx:
                add_diff(bot, dx)
    return time.time() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=128)  # per GPU
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    # bf16 = mixed precision (bf16 MFMA, fp32 accumulation/storage) — a
    # SECOND reported line; the BASELINE contract metric stays f32
    ap.add_argument("--dtype", choices=["f32", "bf16"], default="f32")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    import caffe_amd as ca

    # CAFFE_BENCH_CPU=1: CI dry-run of THIS script's multi-rank plumbing
    # (gloo rendezvous, barriers, max-over-ranks timing, JSON emission) on
    # CPU with no RCCL — tests/test_bench_dryrun.py runs it at world 2.
    # Never used for reported numbers.
    cpu_dry = os.environ.get("CAFFE_BENCH_CPU") == "1"
    if cpu_dry:
        ca.set_mode("cpu")
    else:
        # initialize HIP through /opt/rocm's runtime BEFORE torch gets a
        # chance to load its bundled one (same soname; first load wins
        # process-wide)
        ca.set_mode("gpu", local_rank)

    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group("gloo", rank=rank, world_size=world)
    shapes = {"resnet50": (3, 224, 224), "alexnet": (3, 227, 227),
              "googlenet": (3, 224, 224)}
    ca.set_synthetic_shape(*shapes[args.model], 1000)
    ca.set_random_seed(1371)
    if args.dtype == "bf16" and not cpu_dry:
        ca.set_compute("bf16")

    import subprocess
    gen = os.path.join(REPO, "models", "generated",
                       f"{args.model}_solver.prototxt")
    if not os.path.exists(gen):
        if rank == 0:
            subprocess.check_call([sys.executable,
                                   os.path.join(REPO, "models",
                                                "gen_models.py")])
        else:  # avoid the generation race: wait for rank 0
            for _ in range(120):
                if os.path.exists(gen):
                    break
                time.sleep(0.5)
    solver = ca.Solver(path=gen, batch_override=args.batch)

    if world > 1 and not cpu_dry:
        # ncclUniqueId from rank 0 over the gloo store (the reference used
        # MPI_Bcast, parallel.cpp:45)
        obj = [solver.comm_unique_id() if rank == 0 else None]
        dist.broadcast_object_list(obj, src=0)
        solver.comm_init(rank, world, obj[0])
        solver.bcast_weights()

    def barrier():
        # local device drained first, THEN the cross-rank barrier: t0/t1 on
        # every rank brackets only finished GPU work (contract ordering)
        ca.device_synchronize()
        if dist:
            dist.barrier()

    solver.step(args.warmup)
    barrier()
    t0 = time.time()
    solver.step(args.steps)
    barrier()
    elapsed = time.time() - t0
    if dist:
        import torch
        t = torch.tensor([elapsed])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    value = world * args.batch * args.steps / elapsed

    # roofline leg: 3 instrumented steps, HIP-event timed per kernel class
    # on the launching stream.  ALL ranks step (the collectives are inside
    # the step — a rank-0-only step would deadlock the communicator).
    ca.perf_reset()
    ca.set_perf_timing(True)
    solver.step(1 if cpu_dry else 3)
    ca.device_synchronize()
    ca.set_perf_timing(False)
    result = None
    if rank == 0:
        perf = ca.perf_snapshot()
        g = {"flops": 0.0, "ns": 0.0, "launches": 0}
        for k, v in perf.items():
            if k.startswith("gemm"):
                g["flops"] += v["flops"]
                g["ns"] += v["ns"]
                g["launches"] += v["launches"]
        achieved = g["flops"] / max(g["ns"], 1) * 1e9  # FLOP/s
        # measured HBM bytes per GEMM-class launch, from the committed PMC
        # capture (tools/pmc_traffic.py, FETCH_SIZE x2-corrected +
        # WRITE_SIZE per MI355X_MICROARCH.md §HBM); null if not captured
        traffic = None
        pmc_path = os.path.join(REPO, "profiles",
                                f"pmc_traffic_{args.model}.json")
        if os.path.exists(pmc_path):
            try:
                pmc = json.load(open(pmc_path))
                gc = pmc["classes"].get("gemm")
                if gc and gc.get("launches"):
                    traffic = ((gc["fetch_bytes"] + gc["write_bytes"])
                               / gc["launches"])
            except Exception:
                traffic = None
        peak = PEAK_BF16_MFMA if args.dtype == "bf16" else PEAK_F32_MFMA
        roofline = {
            "bound": "mfma",
            "achieved": achieved,
            "peak": peak,
            "unit": "TFLOP/s",
            "frac": achieved / peak,
            "traffic": traffic,
            "traffic_note": ("HBM bytes/launch of the GEMM class, "
                            "profiles/pmc_traffic_*.json "
                            "(FETCH_SIZE x2 + WRITE_SIZE)")
                           if traffic is not None else None,
            "kernel": "gemm_f32 (conv/IP contractions)",
            "kernel_time_frac": None,
        }
        tot_ns = sum(v["ns"] for v in perf.values())
        if tot_ns > 0:
            roofline["kernel_time_frac"] = g["ns"] / tot_ns
        roofline["classes"] = {
            k: {"ms": v["ns"] / 1e6,
                "tf": v["flops"] / max(v["ns"], 1) / 1e3}
            for k, v in sorted(perf.items(), key=lambda kv: -kv[1]["ns"])}

        cpu_baseline = None
        # the committed oracle walker restates ResNet-50 only; pairing it
        # with another model's metric would be wrong — omit instead
        if world == 1 and not args.no_cpu_baseline and \
                args.model == "resnet50" and not cpu_dry:
            bcpu = 4
            t_cpu = oracle_resnet50_step(bcpu)
            cpu_baseline = {
                "value": bcpu / t_cpu,
                "unit": "images/sec",
                "cores": int(os.environ.get("OMP_NUM_THREADS",
                                            os.cpu_count())),
                "kind": "port",
                "sample": f"1 fwd+bwd+update step at batch {bcpu} "
                          f"({t_cpu:.1f}s)",
            }

        result = {
            "metric": "images/sec",
            "value": value,
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "workload": f"{args.model} {args.dtype} training, "
                            f"per-GPU batch "
                            f"{args.batch}, synthetic "
                            f"{'x'.join(map(str, shapes[args.model]))}",
                "model": args.model,
                "global_batch": world * args.batch,
                "parallelism": f"dp{world}",
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        # reference-style summary lines (solver.cpp:619-628, parallel.cpp:85)
        per_gpu = value / world
        print(f"Solver performance on device 0: "
              f"{per_gpu / args.batch:.4g} * {args.batch} = "
              f"{per_gpu:.5g} img/sec", file=sys.stderr)
        print(f"Overall multi-GPU performance: {value:.5g} img/sec",
              file=sys.stderr)
        print(json.dumps(result))
    if dist:
        dist.destroy_process_group()
    return result


if __name__ == "__main__":
    main()
