#!/usr/bin/env python3
"""Measure frozen vs TreeSA-refined paths on the GPU (re-freeze decision
evidence for VERDICT r01 item 7): same fixture network, two contraction
paths, per-path metric GFLOP/s AND wall per contraction. The refinement is
deterministic (seeded chained restarts, same recipe as the r02 CPU sweep).

Usage: python scripts/refined_trial.py [rqc36|syc49] [steps]
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnc_amd.contraction_path import ContractionPath, flatten_network
from tnc_amd.executor import ContractionEngine, arena_bytes, plan_steps
from tnc_amd.fixtures import FIXTURES, load_fixture
from tnc_amd.tensor import LeafTensor
from tnc_amd.treesa import refine_replace_path


def main():
    fx = sys.argv[1] if len(sys.argv) > 1 else "rqc36"
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 3
    tn, rp, meta = load_fixture(fx)
    cap = FIXTURES[fx].get("size_cap")
    dtype = meta.get("dtype", "c128")
    esize = 8 if dtype == "c64" else 16

    leaves_v = [LeafTensor(t.legs, t.bond_dims) for t in tn.tensors]
    t0 = time.time()
    best_top, best_op = rp, meta["op_cost"]
    for r in range(6):
        top, op, peak = refine_replace_path(leaves_v, best_top, moves=200_000,
                                            seed=100 + r, size_cap=cap)
        if op < best_op:
            best_top, best_op = top, op
    refine_s = time.time() - t0

    for label, top in (("frozen", rp), ("refined", best_top)):
        replace = ContractionPath.simple(top)
        leaves, fsteps, _ = flatten_network(tn, replace)
        infos = plan_steps(leaves, fsteps)
        ab = arena_bytes(leaves, fsteps, infos, esize)
        mf = sum(i.flops for i in infos)
        if ab > 250e9:
            print(json.dumps({"fixture": fx, "path": label,
                              "skipped": f"arena {ab/1e9:.0f} GB"}))
            continue
        eng = ContractionEngine(tn, replace, dtype=dtype)
        eng.contract()  # warmup / capture
        t0 = time.perf_counter()
        for _ in range(steps):
            eng.contract()
        wall = (time.perf_counter() - t0) / steps
        eng.close()
        print(json.dumps({
            "fixture": fx, "path": label, "metric_flops": mf,
            "arena_gb": ab / 1e9, "ms_per_contraction": wall * 1e3,
            "gflops": mf / wall / 1e9,
            "refine_s": refine_s if label == "refined" else 0.0,
        }), flush=True)


if __name__ == "__main__":
    main()
