#!/usr/bin/env python3
"""Full-size VALUE parity for syc49 (config 5): contract the frozen fixture
on the GPU (c64 path) AND in full through the oracle on the host (c128),
print both amplitudes and the relative error.

Not part of the pytest gpu suite: the oracle walk takes ~5-10 minutes on
the box's host cores (~3.7e14 flops at host-BLAS rate, peak live ~150 GB
in c128), which would dominate the suite's time budget. Run on a lease;
the output is committed under profiles/ (see syc49_full_parity_r02.md).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("OPENBLAS_NUM_THREADS", "64")
os.environ.setdefault("OMP_NUM_THREADS", "64")


def main():
    from oracle import contract_network
    from oracle.adapters import network_to_otensors
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture

    tn, rp, meta = load_fixture("syc49")
    assert meta.get("dtype") == "c64"
    replace = ContractionPath.simple(rp)

    eng = ContractionEngine(tn, replace, dtype="c64")
    t0 = time.time()
    eng.contract()
    legs, data = eng.result()
    eng.close()
    gpu = complex(data)
    print(f"[syc49-parity] GPU (c64): {gpu!r}  ({time.time()-t0:.1f}s)",
          flush=True)

    t0 = time.time()
    ref = contract_network(network_to_otensors(tn), rp)
    oracle_amp = complex(ref.data)
    print(f"[syc49-parity] oracle (c128): {oracle_amp!r}  "
          f"({time.time()-t0:.1f}s)", flush=True)
    rel = abs(gpu - oracle_amp) / abs(oracle_amp)
    print(f"[syc49-parity] relative error: {rel:.3e} "
          f"(c64 path vs c128 oracle; f32 accumulation over 3.7e14 flops)",
          flush=True)


if __name__ == "__main__":
    main()
