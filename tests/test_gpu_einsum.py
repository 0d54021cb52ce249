"""GPU einsum parity vs the oracle through the C ABI (tn_einsum_c128).

Covers every kernel class: smallk (gate-apply shapes), anyk (skinny),
dot (scalar out), outer product (K=1), MFMA GEMM, ragged GEMM (v1), packed
and pack-free operands, unpack (caller-chosen out order), strided views,
pow2 and non-pow2 dims.
"""

import numpy as np
import pytest

import oracle
from oracle import OTensor
from oracle.core import symmetric_difference

pytestmark = pytest.mark.gpu


def _rand(shape, rng):
    return (rng.standard_normal(shape) + 1j * rng.standard_normal(shape)).astype(
        np.complex128
    )


def run_case(a_labels, a_shape, b_labels, b_shape, seed=0, out_labels=None,
             rtol=1e-12):
    from tnc_amd import hiplib

    rng = np.random.default_rng(seed)
    a = _rand(a_shape, rng)
    b = _rand(b_shape, rng)
    if out_labels is None:
        dims_a, dims_b = list(a.shape), list(b.shape)
        out_labels, _ = symmetric_difference(a_labels, dims_a, b_labels, dims_b)
    ref = oracle.contract_ndarrays(out_labels, a_labels, a, b_labels, b)
    got = hiplib.einsum_c128(out_labels, a_labels, a, b_labels, b)
    np.testing.assert_allclose(got, ref, rtol=rtol, atol=1e-10)
    return got


def test_golden_pairwise():
    """The reference's own golden vectors through the GPU path."""
    import os

    from tnc_amd import hiplib

    z = np.load(os.path.join(os.path.dirname(__file__), "golden",
                             "contraction_ref.npz"))
    B, A = z["B_data"], z["A_data"]
    bl, al = z["B_legs"].tolist(), z["A_legs"].tolist()
    out_labels, _ = symmetric_difference(bl, B.shape, al, A.shape)
    got = hiplib.einsum_c128(out_labels, bl, B, al, A)
    assert out_labels == z["AxB_legs"].tolist()
    np.testing.assert_allclose(got, z["AxB_data"], atol=1e-14)

    C = z["C_data"]
    cl = z["C_legs"].tolist()
    out_labels, _ = symmetric_difference(cl, C.shape, bl, B.shape)
    got = hiplib.einsum_c128(out_labels, cl, C, bl, B)
    assert out_labels == z["BxC_legs"].tolist()
    np.testing.assert_allclose(got, z["BxC_data"], atol=1e-14)


def test_smallk_pow2():
    # gate application: A = big state (rank 8), B = 2q gate
    run_case([0, 1, 2, 3, 4, 5, 6, 7], [2] * 8, [10, 11, 2, 5], [2] * 4)


def test_smallk_nonpow2():
    run_case([0, 1, 2], [3, 5, 7], [2, 3], [7, 4])


def test_outer_product():
    run_case([0, 1], [4, 5], [2], [6])
    run_case([0], [3], [1], [2])


def test_scalar_dot_small():
    run_case([0, 1], [2, 2], [0, 1], [2, 2])


def test_scalar_dot_large():
    # K = 2^18 -> dot kernel
    run_case(list(range(18)), [2] * 18, list(range(18)), [2] * 18)


def test_rank0_operand():
    from tnc_amd import hiplib

    rng = np.random.default_rng(3)
    a = _rand((), rng)
    b = _rand((2, 3), rng)
    ref = oracle.contract_ndarrays([7, 8], [], a, [7, 8], b)
    got = hiplib.einsum_c128([7, 8], [], a, [7, 8], b)
    np.testing.assert_allclose(got, ref, rtol=1e-12)


def test_gemm_mfma_pow2():
    # M = N = K = 256, pack-free A ([M..,K..] already) and packed B
    run_case([0, 1, 2, 3], [16, 16, 16, 16], [4, 5, 2, 3], [16, 16, 16, 16])


def test_gemm_mfma_qubit_legs():
    # all dims 2: M=2^9, N=2^8, K=2^9
    a_labels = list(range(18))          # 9 M legs + 9 K legs
    b_labels = list(range(9, 18)) + list(range(100, 108))  # 9 K + 8 N
    run_case(a_labels, [2] * 18, b_labels, [2] * 17, rtol=1e-11)


def test_gemm_ragged_v1():
    # non-pow2 dims -> v1 kernel with bounds handling
    run_case([0, 1], [100, 9], [1, 2], [9, 75])
    run_case([0, 1], [67, 130], [1, 2], [130, 41])


def test_gemm_needs_packing_both():
    # A legs ordered K-first, B legs N-first -> both packs run
    run_case([2, 3, 0, 1], [8, 8, 16, 16], [5, 2, 3, 4], [16, 8, 8, 16])


def test_gemm_unpack_interleaved_out():
    # caller requests interleaved out order -> unpack permute
    run_case([0, 1, 2], [32, 8, 64], [2, 3, 4], [64, 8, 32],
             out_labels=[0, 3, 1, 4])


def test_strided_views():
    from tnc_amd import hiplib

    rng = np.random.default_rng(11)
    a = _rand((8, 12, 6), rng).transpose(2, 0, 1)  # non-contiguous view
    b = _rand((12, 10), rng)
    # symdiff out order: A-only [2, 0] then B-only [1]
    ref = oracle.contract_ndarrays([2, 0, 1], [2, 0, 3], a, [3, 1], b)
    got = hiplib.einsum_c128([2, 0, 1], [2, 0, 3], a, [3, 1], b)
    np.testing.assert_allclose(got, ref, rtol=1e-12)
    # strided through the GEMM path too
    a2 = _rand((128, 96), rng).T  # labels [0, 1], shape (96, 128), strided
    b2 = _rand((128, 80), rng)
    ref = oracle.contract_ndarrays([0, 2], [0, 1], a2, [1, 2], b2)
    got = hiplib.einsum_c128([0, 2], [0, 1], a2, [1, 2], b2)
    np.testing.assert_allclose(got, ref, rtol=1e-11)


def test_trace_rejected():
    """Labels absent from both out and the other operand are outside the
    tensor_mult boundary contract -> loud error, not silent garbage."""
    from tnc_amd import hiplib

    rng = np.random.default_rng(12)
    a = _rand((4, 3), rng)
    b = _rand((3, 5), rng)
    with pytest.raises(RuntimeError, match="neither contracted nor in out"):
        hiplib.einsum_c128([2], [0, 1], a, [1, 2], b)  # leg 0 traced


def test_skinny_anyk():
    # M=4, N=2^12, K=2^10 -> anyk kernel (skinny, K>64)
    run_case([0, 1] + list(range(10, 20)), [2, 2] + [2] * 10,
             list(range(10, 20)) + list(range(30, 42)), [2] * 22)


def test_high_rank():
    # rank-24 output, dims 2
    a_labels = list(range(16))
    b_labels = list(range(8, 24))
    run_case(a_labels, [2] * 16, b_labels, [2] * 16, rtol=1e-11)


def test_gemm_splitk():
    # few tiles + deep K -> split-K partials + reduce
    run_case([0, 1], [64, 4096], [1, 2], [4096, 64], rtol=1e-11)
    run_case([0, 1], [128, 65536], [1, 2], [65536, 128], rtol=1e-10)
    # deep-K with multiple qubit legs (pow2 maps + packing)
    a_labels = [0, 1] + list(range(10, 22))
    b_labels = list(range(10, 22)) + [2, 3]
    run_case(a_labels, [2] * 14, b_labels, [2] * 14, rtol=1e-10)


def _run_c64(a_labels, a_shape, b_labels, b_shape, seed=0, rtol=2e-3):
    from tnc_amd import hiplib

    rng = np.random.default_rng(seed)
    a = _rand(a_shape, rng).astype(np.complex64)
    b = _rand(b_shape, rng).astype(np.complex64)
    out_labels, _ = symmetric_difference(a_labels, list(a.shape),
                                         b_labels, list(b.shape))
    ref = oracle.contract_ndarrays(out_labels, a_labels,
                                   a.astype(np.complex128), b_labels,
                                   b.astype(np.complex128))
    got = hiplib.einsum_c64(out_labels, a_labels, a, b_labels, b)
    assert got.dtype == np.complex64
    np.testing.assert_allclose(got, ref, rtol=rtol, atol=1e-4)


def test_dot_tiled_bitperm():
    # K = 2^20 with scrambled leg orders -> tiled bit-permutation dot
    rng = np.random.default_rng(7)
    legs = list(range(20))
    blabels = [legs[p] for p in rng.permutation(20)]
    run_case(legs, [2] * 20, blabels, [2] * 20, rtol=1e-9)


def test_dot_tiled_mixed_pow2_dims():
    # pow2 dims > 2 decompose into bits; K = 2^21 -> tiled path
    run_case([0, 1, 2, 3, 4, 5, 6], [4, 8, 2, 16, 4, 64, 8],
             [6, 3, 0, 2, 5, 4, 1], [8, 16, 4, 2, 64, 4, 8], rtol=1e-9)


def test_c64_smallk():
    _run_c64([0, 1, 2, 3, 4, 5], [2] * 6, [9, 10, 2, 4], [2] * 4)


def test_c64_gemm_mfma():
    # f32 MFMA path, M=N=K=256 with packing
    _run_c64([0, 1, 2, 3], [16, 16, 16, 16], [4, 5, 2, 3], [16, 16, 16, 16])


def test_c64_gemm_splitk():
    _run_c64([0, 1], [64, 8192], [1, 2], [8192, 64], rtol=5e-3)


def test_c64_dot():
    _run_c64(list(range(16)), [2] * 16, list(range(16)), [2] * 16, rtol=5e-3)


def test_c64_network_engine():
    """Full network in c64 vs the c128 oracle (config-5 precision trade)."""
    from tnc_amd import Greedy
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.executor import ContractionEngine

    tn = random_circuit(14, 8, 0.5, 0.5, 5, ConnectivityLayout.EAGLE)
    replace = Greedy().find_path(tn).replace_path()
    eng = ContractionEngine(tn, replace, dtype="c64")
    try:
        eng.contract()
        legs, data = eng.result()
        assert data.dtype == np.complex64
    finally:
        eng.close()
    from oracle import contract_network
    from oracle.adapters import network_to_otensors

    ref = contract_network(network_to_otensors(tn), replace)
    np.testing.assert_allclose(data, ref.data, rtol=1e-3, atol=1e-5)


def test_fuzz_random_einsums():
    """30 seeded random (labels, dims) configurations across the dispatch
    space: shared/unshared legs in arbitrary interleavings, dims 1..8
    (pow2 and not), rank 0..6 — each vs the oracle."""
    rng = np.random.default_rng(123)
    for case in range(30):
        ra = int(rng.integers(0, 7))
        rb = int(rng.integers(0, 7))
        pool = list(range(14))  # ra + rb-only <= 12 < 14, so choices never run dry
        a_labels = list(rng.choice(pool, size=ra, replace=False))
        # share a random subset of A's labels, in shuffled positions
        nshare = int(rng.integers(0, min(ra, rb) + 1)) if min(ra, rb) else 0
        shared = list(rng.choice(a_labels, size=nshare, replace=False)) \
            if nshare else []
        b_only = [l for l in rng.choice(
            [p for p in pool if p not in a_labels],
            size=rb - nshare, replace=False)] if rb - nshare else []
        b_labels = shared + b_only
        rng.shuffle(b_labels)
        dims = {l: int(rng.choice([1, 2, 3, 4, 8])) for l in pool}
        a_shape = [dims[l] for l in a_labels]
        b_shape = [dims[l] for l in b_labels]
        run_case(a_labels, a_shape, b_labels, b_shape,
                 seed=1000 + case, rtol=1e-10)


def test_fuzz_random_einsums_c64():
    """12 seeded random configurations through the c64 path."""
    from tnc_amd import hiplib

    rng = np.random.default_rng(77)
    for case in range(12):
        ra = int(rng.integers(1, 6))
        rb = int(rng.integers(1, 6))
        pool = list(range(12))
        a_labels = list(rng.choice(pool, size=ra, replace=False))
        nshare = int(rng.integers(0, min(ra, rb) + 1))
        shared = list(rng.choice(a_labels, size=nshare, replace=False)) \
            if nshare else []
        b_only = [l for l in rng.choice(
            [p for p in pool if p not in a_labels],
            size=rb - nshare, replace=False)] if rb - nshare else []
        b_labels = shared + b_only
        rng.shuffle(b_labels)
        dims = {l: int(rng.choice([1, 2, 4, 8])) for l in pool}
        r2 = np.random.default_rng(2000 + case)
        a = _rand([dims[l] for l in a_labels], r2).astype(np.complex64)
        b = _rand([dims[l] for l in b_labels], r2).astype(np.complex64)
        out_labels, _ = symmetric_difference(a_labels, list(a.shape),
                                             b_labels, list(b.shape))
        ref = oracle.contract_ndarrays(out_labels, a_labels,
                                       a.astype(np.complex128), b_labels,
                                       b.astype(np.complex128))
        got = hiplib.einsum_c64(out_labels, a_labels, a, b_labels, b)
        np.testing.assert_allclose(got, ref, rtol=2e-3, atol=1e-4)


# ---------------------------------------------------------------------------
# Bit-exact index permutations (north star: "bit-exact for index
# permutations"). Multiplication by exactly 1.0+0.0j and accumulation of
# exact zeros are lossless in IEEE754, so any einsum that is semantically a
# permutation must reproduce the oracle's transpose EXACTLY — through the
# gather kernels (scalar B) and through the full TTGT pack/GEMM/unpack
# machinery (identity B), pow2 and non-pow2 dims.
# ---------------------------------------------------------------------------


def _exact_permute_scalar_b(a_labels, a_shape, out_labels, seed, c64=False):
    """A x scalar-1 with a caller-chosen out order == np.transpose, exactly."""
    from tnc_amd import hiplib

    rng = np.random.default_rng(seed)
    a = _rand(a_shape, rng)
    if c64:
        a = a.astype(np.complex64)
    one = np.ones((), dtype=a.dtype)
    fn = hiplib.einsum_c64 if c64 else hiplib.einsum_c128
    got = fn(out_labels, a_labels, a, [], one)
    perm = [a_labels.index(l) for l in out_labels]
    expect = np.transpose(a, perm)
    np.testing.assert_array_equal(got, expect)


def test_permute_scalar_b_exact_nonpow2():
    _exact_permute_scalar_b([0, 1, 2, 3], [3, 5, 7, 2], [2, 0, 3, 1], seed=11)


def test_permute_scalar_b_exact_pow2():
    _exact_permute_scalar_b(list(range(10)), [2] * 10,
                            [9, 0, 7, 1, 5, 2, 6, 3, 8, 4], seed=12)


def test_permute_scalar_b_exact_c64():
    _exact_permute_scalar_b([0, 1, 2], [6, 10, 9], [2, 0, 1], seed=13,
                            c64=True)


def _exact_permute_identity_b(m_dims, k_dims, seed, c64=False):
    """A contracted with an identity over its K legs is a pure permutation:
    the TTGT path (pack permutes k_permute_ct/k_permute_tile, MFMA GEMM,
    split-K, unpack) must reproduce A's transpose bit-for-bit."""
    from tnc_amd import hiplib

    rng = np.random.default_rng(seed)
    nm, nk = len(m_dims), len(k_dims)
    m_labels = list(range(nm))
    k_labels = [100 + i for i in range(nk)]
    n_labels = [200 + i for i in range(nk)]
    # interleave K legs into A's storage order to force a pack permute
    a_labels, a_dims = [], []
    for i in range(max(nm, nk)):
        if i < nk:
            a_labels.append(k_labels[i])
            a_dims.append(k_dims[i])
        if i < nm:
            a_labels.append(m_labels[i])
            a_dims.append(m_dims[i])
    a = _rand(a_dims, rng)
    K = int(np.prod(k_dims))
    b = np.eye(K, dtype=np.complex128).reshape(tuple(k_dims) + tuple(k_dims))
    b_labels = k_labels + n_labels
    if c64:
        a = a.astype(np.complex64)
        b = b.astype(np.complex64)
    # interleaved out order to force an unpack as well
    out_labels = []
    for i in range(max(nm, nk)):
        if i < nk:
            out_labels.append(n_labels[nk - 1 - i])
        if i < nm:
            out_labels.append(m_labels[i])
    fn = hiplib.einsum_c64 if c64 else hiplib.einsum_c128
    got = fn(out_labels, a_labels, a, b_labels, b)
    # expected: A with K legs relabeled to N, transposed to the out order
    relabel = dict(zip(k_labels, n_labels))
    a_as_out = [relabel.get(l, l) for l in a_labels]
    perm = [a_as_out.index(l) for l in out_labels]
    expect = np.transpose(a, perm)
    np.testing.assert_array_equal(got, expect)


def test_permute_identity_b_gemm_exact_nonpow2():
    # m = 296, n = k = 72: GEMM-worthy (k>=16, m>=128, n>=64), ragged dims
    _exact_permute_identity_b([37, 8], [9, 8], seed=21)


def test_permute_identity_b_gemm_exact_pow2():
    # all dims 2, m = 256, n = k = 256: the bit-permutation tiled path
    _exact_permute_identity_b([2] * 8, [2] * 8, seed=22)


def test_permute_identity_b_gemm_exact_c64():
    _exact_permute_identity_b([37, 8], [9, 8], seed=23, c64=True)


def test_final_permutor_exact():
    """The final Permutor (circuit_builder.rs:77-122) is an exact transpose."""
    from tnc_amd import Permutor

    rng = np.random.default_rng(31)
    data = _rand([2, 3, 4, 5], rng)
    legs = [7, 3, 9, 1]
    p = Permutor([1, 9, 3, 7])
    new_legs, new_dims, out = p.apply(legs, list(data.shape), data)
    assert new_legs == [1, 9, 3, 7]
    np.testing.assert_array_equal(out, np.transpose(data, [3, 2, 1, 0]))
