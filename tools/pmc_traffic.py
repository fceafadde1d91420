"""HBM traffic capture for roofline.traffic (run ON the GPU box).

Two separate rocprofv3 --pmc passes (FETCH_SIZE costs 3 TCC slots,
WRITE_SIZE 2 — they cannot share one pass, MI355X_MICROARCH.md §rocprofv3
PMC slots) over a short bench run, plus the gfx950 correction: FETCH_SIZE
reports HALF the bytes of wide coalesced streaming reads (16 B/lane), so
fetch bytes are doubled before use (§HBM).  Output:

  profiles/pmc_traffic_<model>.json   — per kernel-class totals:
      {class: {launches, fetch_bytes (x2-corrected), write_bytes}}
  gpurun_out/pmc_traffic_<model>.txt  — human-readable table

bench.py embeds roofline.traffic (bytes per launch of the dominant GEMM
class) from the committed JSON; re-run this tool whenever kernels change.

Usage (on the GPU box, from the repo root):
  python tools/pmc_traffic.py [model] [steps]
"""
import csv
import glob
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# kernel-name prefix -> perf class (keep in sync with csrc perf classes)
CLASSES = [
    ("k_gemm_f32", "gemm"),
    ("k_gemm_slim", "gemm"),
    ("k_gemm_bf16", "gemm"),
    ("k_gemm2", "gemm"),
    ("k_splitk_reduce", "gemm"),      # part of the split-K GEMM episode
    ("k_weight_flip", "gemm"),
    ("k_bn_", "bn"),
    ("k_im2col", "im2col"),
    ("k_col2im", "col2im"),
    ("k_relu", "relu"),
    ("k_pool", "pool"),
    ("k_lrn", "lrn"),
    ("k_sgd", "sgd"),
    ("k_bias_grad", "reduce"),
    ("k_colsum", "reduce"),
    ("k_add3", "eltwise"),
    ("k_axpy", "eltwise"),
    ("k_axpby", "eltwise"),
    ("k_copy", "eltwise"),
    ("k_set", "eltwise"),
    ("k_acc", "eltwise"),
    ("k_concat", "concat"),
    ("k_dropout", "dropout"),
    ("k_softmax", "softmax"),
    ("k_sm_", "softmax"),
    ("k_fill", "data"),
]


def classify(kname):
    for pfx, cls in CLASSES:
        if pfx in kname:
            return cls
    return "other"


def run_pass(counter, model, steps, tag):
    out = f"/tmp/pmc_{tag}"
    subprocess.run(["rm", "-rf", out])
    cmd = ["rocprofv3", "--pmc", counter, "-d", out, "-o", tag,
           "--output-format", "csv", "--",
           sys.executable, os.path.join(REPO, "bench.py"),
           "--model", model, "--steps", str(steps), "--warmup", "1",
           "--no-cpu-baseline"]
    env = dict(os.environ, TMPDIR="/tmp")
    r = subprocess.run(cmd, cwd="/tmp", env=env, capture_output=True,
                       text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout[-2000:] + r.stderr[-2000:])
        raise SystemExit(f"rocprofv3 {counter} pass failed")
    files = glob.glob(os.path.join(out, "**", "*counter_collection.csv"),
                      recursive=True)
    if not files:
        files = glob.glob(os.path.join(out, "**", "*.csv"), recursive=True)
    assert files, f"no counter csv under {out}"
    per_kernel = {}  # name -> [launches, bytes]
    for f in files:
        with open(f) as fh:
            for row in csv.DictReader(fh):
                name = row.get("Kernel_Name") or row.get("kernel_name", "")
                cname = (row.get("Counter_Name") or
                         row.get("counter_name", ""))
                if counter not in cname:
                    continue
                val = float(row.get("Counter_Value") or
                            row.get("counter_value", 0))
                e = per_kernel.setdefault(name, [0, 0.0])
                e[0] += 1
                e[1] += val
    return per_kernel


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "resnet50"
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 2
    fetch = run_pass("FETCH_SIZE", model, steps, f"f_{model}")
    write = run_pass("WRITE_SIZE", model, steps, f"w_{model}")

    classes = {}
    for name, (n, kb) in fetch.items():
        c = classes.setdefault(classify(name), {"launches": 0,
                                                "fetch_bytes": 0.0,
                                                "write_bytes": 0.0})
        c["launches"] += n
        # gfx950: FETCH_SIZE = half of wide-coalesced read bytes -> x2
        c["fetch_bytes"] += kb * 1024.0 * 2.0
    for name, (n, kb) in write.items():
        c = classes.setdefault(classify(name), {"launches": 0,
                                                "fetch_bytes": 0.0,
                                                "write_bytes": 0.0})
        c["write_bytes"] += kb * 1024.0

    result = {
        "model": model,
        "steps_counted": "bench --steps %d --warmup 1 (full process incl. "
                         "warmup + iter-0 test pass)" % steps,
        "correction": "FETCH_SIZE x2 (gfx950 halves wide coalesced reads, "
                      "MI355X_MICROARCH.md §HBM); WRITE_SIZE raw",
        "classes": classes,
    }
    # written under gpurun_out/ (the only dir that travels back from the
    # GPU box); copy into profiles/ and commit on the dev side
    os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
    jpath = os.path.join(REPO, "gpurun_out", f"pmc_traffic_{model}.json")
    with open(jpath, "w") as f:
        json.dump(result, f, indent=1, sort_keys=True)
    os.makedirs(os.path.join(REPO, "gpurun_out"), exist_ok=True)
    tpath = os.path.join(REPO, "gpurun_out", f"pmc_traffic_{model}.txt")
    with open(tpath, "w") as f:
        f.write(f"# {model}: per-class HBM bytes (fetch x2-corrected), "
                f"{steps} bench steps + warmup\n")
        for cls, c in sorted(classes.items(),
                             key=lambda kv: -kv[1]["fetch_bytes"]):
            f.write(f"{cls:10s} launches={c['launches']:6d} "
                    f"fetchGB={c['fetch_bytes'] / 1e9:9.2f} "
                    f"writeGB={c['write_bytes'] / 1e9:9.2f}\n")
    print(f"wrote {jpath} and {tpath}")


if __name__ == "__main__":
    main()
