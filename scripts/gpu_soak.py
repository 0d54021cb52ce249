#!/usr/bin/env python3
"""20-random-circuit GPU parity soak vs the oracle (run on a lease)."""
import sys
sys.path.insert(0, ".")
import numpy as np
from oracle import contract_network
from oracle.adapters import network_to_otensors
from tnc_amd import RandomGreedy
from tnc_amd.builders import random_circuit
from tnc_amd.connectivity import ConnectivityLayout
from tnc_amd.executor import ContractionEngine

ok = 0
for case in range(20):
    q = 10 + case % 7
    r = 6 + (case * 3) % 8
    p1 = 0.3 + 0.05 * (case % 9)
    p2 = 0.4 + 0.05 * (case % 9)
    tn = random_circuit(q, r, p1, p2, 500 + case, ConnectivityLayout.EAGLE)
    replace = RandomGreedy(4, seed=case).find_path(tn).replace_path()
    ref = contract_network(network_to_otensors(tn), replace)
    eng = ContractionEngine(tn, replace)
    try:
        eng.contract()
        _, got = eng.result()
    finally:
        eng.close()
    np.testing.assert_allclose(got, ref.data, rtol=1e-10, atol=1e-12,
                               err_msg=f"case {case} q={q} r={r}")
    ok += 1
print(f"SOAK OK: {ok}/20 random circuits match the oracle at 1e-10")
