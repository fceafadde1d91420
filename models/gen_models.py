#!/usr/bin/env python3
"""Re-emit the four reference model/solver prototxts from compact JSON specs.

The specs under models/specs/*.json were extracted (as data) from the
reference's own prototxt files (models/{resnet50,bvlc_alexnet,bvlc_googlenet}/
{train_val,solver}.prototxt and examples/mnist/lenet_{train_test,solver}.prototxt
in Caffe-MPI).  This generator re-serialises them into protobuf text format so
the engine consumes byte-equivalent *semantics* (same layers, same params) —
the "models run unmodified" contract of BASELINE.json.

Usage: python3 models/gen_models.py [outdir]   (default models/generated)
"""
import json
import os
import sys

SPECS = os.path.join(os.path.dirname(os.path.abspath(__file__)), "specs")

# Fields whose values are strings in protobuf text format (need quotes).
STRING_FIELDS = {
    "name", "type", "bottom", "top", "source", "mean_file", "net",
    "snapshot_prefix", "lr_policy", "regularization_type", "momentum_policy",
    "train_net", "test_net",
}
# Fields whose values are bare enum identifiers (no quotes).
ENUM_FIELDS = {
    "pool", "backend", "phase", "operation", "solver_mode", "norm_region",
    "variance_norm", "solver_type",
}


def emit(msg, indent=0):
    out = []
    pad = "  " * indent
    for k, v in msg:
        if isinstance(v, list) and v and isinstance(v[0], list):
            out.append(f"{pad}{k} {{")
            out.append(emit(v, indent + 1))
            out.append(f"{pad}}}")
        else:
            if k in STRING_FIELDS:
                out.append(f'{pad}{k}: "{v}"')
            elif k in ENUM_FIELDS:
                out.append(f"{pad}{k}: {v}")
            else:
                out.append(f"{pad}{k}: {v}")
    return "\n".join(out)


def filler_type_is_string(msg):
    # weight_filler { type: "xavier" } — 'type' inside filler is a string; the
    # STRING_FIELDS rule already covers it.
    return True


def generate(outdir):
    os.makedirs(outdir, exist_ok=True)
    names = ["lenet", "alexnet", "resnet50", "googlenet"]
    for n in names:
        spec = json.load(open(os.path.join(SPECS, f"{n}.json")))
        with open(os.path.join(outdir, f"{n}_train_val.prototxt"), "w") as f:
            f.write(emit(spec) + "\n")
        sspec = json.load(open(os.path.join(SPECS, f"{n}_solver.json")))
        # point the solver at the generated net file
        sspec = [["net", os.path.join(outdir, f"{n}_train_val.prototxt")]
                 if k == "net" else [k, v] for k, v in sspec]
        with open(os.path.join(outdir, f"{n}_solver.prototxt"), "w") as f:
            f.write(emit(sspec) + "\n")
    return [os.path.join(outdir, f"{n}_train_val.prototxt") for n in names]


if __name__ == "__main__":
    outdir = sys.argv[1] if len(sys.argv) > 1 else os.path.join(
        os.path.dirname(os.path.abspath(__file__)), "generated")
    for p in generate(outdir):
        print(p)
