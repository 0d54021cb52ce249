#!/usr/bin/env python3
"""Per-step timing breakdown of the fixture contraction (run on the GPU)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tnc_amd.contraction_path import ContractionPath
from tnc_amd.executor import ContractionEngine
from tnc_amd.fixtures import load_fixture

KIND = {0: "smallk", 1: "dot", 2: "gemm", 3: "gemm+unpack", 5: "dot-lin",
        6: "dot-tile"}


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "rqc36"
    tn, rp, meta = load_fixture(name)
    eng = ContractionEngine(tn, ContractionPath.simple(rp),
                            dtype=meta.get("dtype", "c128"))
    eng.contract()  # warmup
    elapsed, step_ms, gemm_ms, kinds = eng.contract_profiled()
    total = sum(step_ms)
    print(f"{name}: {len(eng.steps)} steps, elapsed {elapsed:.1f} ms, "
          f"sum(step_ms) {total:.1f} ms, flops {eng.total_flops:.3e}")
    agg = {}
    for s, info in enumerate(eng.infos):
        k = KIND[kinds[s]]
        a = agg.setdefault(k, [0.0, 0.0, 0])
        a[0] += step_ms[s]
        a[1] += info.flops
        a[2] += 1
    for k, (ms, fl, n) in sorted(agg.items(), key=lambda x: -x[1][0]):
        print(f"  {k:12s} n={n:4d} ms={ms:9.1f} flops={fl:.3e} "
              f"-> {fl/ms/1e9 if ms else 0:8.1f} GF/s")
    order = sorted(range(len(step_ms)), key=lambda s: -step_ms[s])[:25]
    print("top steps:")
    for s in order:
        i = eng.infos[s]
        print(f"  step {s:4d} {KIND[kinds[s]]:12s} ms={step_ms[s]:9.2f} "
              f"gemm_ms={gemm_ms[s]:9.2f} M={int(i.m):>9} N={int(i.n):>7} "
              f"K={int(i.k):>7} {i.flops/step_ms[s]/1e9:9.1f} GF/s")
    eng.close()


if __name__ == "__main__":
    main()
