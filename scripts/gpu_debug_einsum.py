#!/usr/bin/env python3
"""GPU einsum debug battery: prints max-diff per case and mismatch structure."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import oracle
from oracle.core import symmetric_difference
from tnc_amd import hiplib


def case(name, a_labels, a, b_labels, b, out_labels=None):
    if out_labels is None:
        out_labels, _ = symmetric_difference(
            a_labels, a.shape, b_labels, b.shape
        )
    ref = oracle.contract_ndarrays(out_labels, a_labels, a, b_labels, b)
    got = hiplib.einsum_c128(out_labels, a_labels, a, b_labels, b)
    diff = np.abs(got - ref)
    denom = np.abs(ref) + 1e-30
    print(f"{name}: maxabs={diff.max():.3e} maxrel={(diff/denom).max():.3e} "
          f"mismatch={np.count_nonzero(diff > 1e-8)}/{diff.size}")
    if diff.max() > 1e-8 and ref.ndim == 2:
        # structure probes for matmul-like cases
        print("   got == ref.T ?", np.allclose(got, ref.T) if ref.shape[0] == ref.shape[1] else "n/a")
        bad = np.argwhere(diff > 1e-8)
        print("   first bad idx:", bad[:5].tolist(), " shape:", ref.shape)
        ok = np.argwhere(diff <= 1e-8)
        print("   first ok idx:", ok[:5].tolist())
    return diff.max()


def r(shape, seed):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal(shape) + 1j * rng.standard_normal(shape))


def main():
    # 1. tiny matmul smallk
    case("tiny-matmul", [0, 1], r((4, 3), 1), [1, 2], r((3, 5), 2))
    # 2. v1 ragged pack-free
    case("v1-ragged", [0, 1], r((100, 9), 3), [1, 2], r((9, 75), 4))
    # 3. v1 ragged with pack (B transposed layout)
    case("v1-ragged-packB", [0, 1], r((100, 9), 3), [2, 1],
         np.ascontiguousarray(r((9, 75), 4).T))
    # 4. mfma 64x64x64 both free
    case("mfma-64", [0, 1], r((64, 64), 5), [1, 2], r((64, 64), 6))
    # 5. mfma 64x64x64 B packed
    case("mfma-64-packB", [0, 1], r((64, 64), 5), [2, 1],
         np.ascontiguousarray(r((64, 64), 6).T))
    # 6. mfma 256^3 both free
    case("mfma-256", [0, 1], r((256, 256), 7), [1, 2], r((256, 256), 8))
    # 7. mfma 128x64 K=64
    case("mfma-128x64", [0, 1], r((128, 64), 9), [1, 2], r((64, 64), 10))
    # 8. mfma 64x128 K=64
    case("mfma-64x128", [0, 1], r((64, 64), 11), [1, 2], r((64, 128), 12))
    # 9. mfma 64x64 K=128
    case("mfma-64x64x128", [0, 1], r((64, 128), 13), [1, 2], r((128, 64), 14))
    # 10. golden B x A
    z = np.load(os.path.join(os.path.dirname(__file__), "..", "tests",
                             "golden", "contraction_ref.npz"))
    case("golden-BxA", z["B_legs"].tolist(), z["B_data"],
         z["A_legs"].tolist(), z["A_data"])
    # 11. strided smallk
    a = np.asarray(r((8, 12, 6), 15)).transpose(2, 0, 1)
    b = np.asarray(r((12, 10), 16))
    case("strided-smallk", [2, 0, 3], a, [3, 1], b)
    # 12. smallk skinny contiguous (golden-like dims)
    case("skinny-748", [0, 1, 2], r((7, 4, 8), 17), [2, 3], r((8, 6), 18))


if __name__ == "__main__":
    main()
