#!/usr/bin/env python3
"""Convert the reference's golden contraction vectors to a committed fixture.

Reads /root/reference/tnc/src/tensornetwork/contraction_test_data.json
(the reference's own golden test data, used by its tests at
tnc/src/tensornetwork/contraction.rs:154-261) and writes
tests/golden/contraction_ref.npz. Run in the dev container (where
/root/reference is mounted); the .npz is committed and travels to GPU boxes.
"""

import json
import os

import numpy as np

SRC = "/root/reference/tnc/src/tensornetwork/contraction_test_data.json"
DST = os.path.join(os.path.dirname(__file__), "..", "tests", "golden", "contraction_ref.npz")


def main():
    with open(SRC) as f:
        raw = json.load(f)
    out = {}
    for name, t in raw.items():
        data = np.array([complex(re, im) for re, im in t["data"]], dtype=np.complex128)
        out[f"{name}_legs"] = np.array(t["legs"], dtype=np.int64)
        out[f"{name}_data"] = data.reshape(t["shape"])
    np.savez_compressed(DST, **out)
    print(f"wrote {DST}: {sorted(raw)}")


if __name__ == "__main__":
    main()
