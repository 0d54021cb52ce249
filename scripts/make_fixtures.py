#!/usr/bin/env python3
"""Freeze the benchmark networks (BASELINE.json configs 2-4) plus their
contraction paths to tests/golden/*.json (the reference benchmark's
sweep/run split, benchmark/src/main.rs:223-242)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnc_amd import PartitionSearch, RandomGreedy
from tnc_amd.cost import contract_cost_tensors
from tnc_amd.fixtures import FIXTURES, build_fixture, fixture_path, save_network
from tnc_amd.tensor import LeafTensor


def metric_flops(tn, replace_toplevel):
    """Sum over steps of (8s-2)*o (contraction_cost.rs:26-32) — the GFLOP/s
    numerator for the frozen path."""
    views = [LeafTensor(t.legs, t.bond_dims) for t in tn.tensors]
    total = 0.0
    for i, j in replace_toplevel:
        total += contract_cost_tensors(views[i], views[j])
        views[i] = views[i] ^ views[j]
    return total


def main():
    for name, cfg in FIXTURES.items():
        tn = build_fixture(name)
        # best-of search (the config-3 path is "the best path
        # found" — RandomGreedy(many) + the partition-guided quality tier
        # standing in for cotengra HyperOptimizer)
        finder_args = dict(ks=(2, 3, 4, 6, 8, 12, 16), seeds=(0, 1, 2, 3),
                           trials=cfg["trials"],
                           size_cap=cfg.get("size_cap"))
        rg = PartitionSearch(**finder_args).find_path(tn)
        replace = rg.replace_path()
        assert not replace.nested
        flops = metric_flops(tn, replace.toplevel)
        meta = {
            "config": {k: str(v) for k, v in cfg.items()},
            "dtype": cfg.get("dtype", "c128"),
            "op_cost": rg.flops,
            "peak_size_elems": rg.size,
            "metric_flops": flops,
            # provenance of the frozen path (surfaced by bench.py's
            # config.path_provenance)
            "path_finder": {
                "finder": "PartitionSearch",
                "ks": list(finder_args["ks"]),
                "seeds": list(finder_args["seeds"]),
                "trials": finder_args["trials"],
                "size_cap": finder_args["size_cap"],
                "greedy_seed": 42,
            },
        }
        save_network(tn, fixture_path(name), replace.toplevel, meta)
        print(
            f"{name}: {len(tn.tensors)} tensors, {len(replace.toplevel)} steps, "
            f"op_cost={rg.flops:.3e}, peak={rg.size:.3e} elems, "
            f"metric_flops={flops:.4e}"
        )


if __name__ == "__main__":
    main()
