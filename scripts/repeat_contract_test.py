#!/usr/bin/env python3
"""Repeated-contraction soak: N contractions per engine, both dtypes, with
result-identity checks — catches state corruption across contract calls."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from tnc_amd import Greedy, RandomGreedy
from tnc_amd.builders import random_circuit
from tnc_amd.connectivity import ConnectivityLayout
from tnc_amd.contraction_path import ContractionPath
from tnc_amd.executor import ContractionEngine
from tnc_amd.fixtures import load_fixture


def soak(tn, replace, dtype, n, label):
    eng = ContractionEngine(tn, replace, dtype=dtype)
    ref = None
    for it in range(n):
        if it == 1:
            eng.contract_profiled()  # mix profiled calls in
        else:
            eng.contract()
        _, data = eng.result()
        if ref is None:
            ref = data
        else:
            assert np.array_equal(ref, data), f"{label}: drift at iter {it}"
    eng.close()
    print(f"{label}: {n} contractions identical OK")


def main():
    reps = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    for dtype in ("c128", "c64"):
        tn = random_circuit(16, 10, 0.5, 0.6, 7, ConnectivityLayout.EAGLE)
        replace = RandomGreedy(8).find_path(tn).replace_path()
        soak(tn, replace, dtype, reps, f"16q-{dtype}")
    # the actual rqc24 fixture, many reps
    tn, rp, _ = load_fixture("rqc24")
    soak(tn, ContractionPath.simple(rp), "c128", reps, "rqc24-c128")
    soak(tn, ContractionPath.simple(rp), "c64", reps, "rqc24-c64")
    print("soak OK")


if __name__ == "__main__":
    main()
