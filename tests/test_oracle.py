"""Oracle parity with the reference's own golden vectors and known-answer
tests (see DESIGN.md "Oracle" for the pinning inventory)."""

import os

import numpy as np
import pytest

import oracle
from oracle import OTensor, contract_network, contract_tensors
from oracle.gates import load_gate, load_gate_adjoint, matrix_adjoint, gate_names

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "contraction_ref.npz")


def load_golden(name):
    z = np.load(GOLDEN)
    return OTensor(z[f"{name}_legs"].tolist(), z[f"{name}_data"])


def test_pairwise_golden():
    """contraction.rs:154-173: B x A and C x B golden tensors, eps 1e-14."""
    A, B, C = load_golden("A"), load_golden("B"), load_golden("C")
    AxB, BxC = load_golden("AxB"), load_golden("BxC")

    out = contract_tensors(B, A)
    assert out.legs == AxB.legs
    np.testing.assert_allclose(out.data, AxB.data, atol=1e-14)

    out = contract_tensors(C, B)
    assert out.legs == BxC.legs
    np.testing.assert_allclose(out.data, BxC.data, atol=1e-14)


def test_path_golden():
    """contraction.rs:175-192: network [A,B,C], path [(1,0),(2,1)] -> ABxC."""
    A, B, C = load_golden("A"), load_golden("B"), load_golden("C")
    ABxC = load_golden("ABxC")
    out = contract_network([A, B, C], [(1, 0), (2, 1)])
    assert out.legs == ABxC.legs
    np.testing.assert_allclose(out.data, ABxC.data, atol=1e-14)


def test_outer_product():
    """contraction.rs:195-229 exact values."""
    t1 = OTensor([0], np.array([1 + 0j, 2 + 5j, 3 - 1j]))
    t2 = OTensor([1], np.array([-4 + 2j, -1j]))
    out = contract_network([t1, t2], [(0, 1)])
    ref = np.array(
        [[-4 + 2j, -1j], [-18 - 16j, 5 - 2j], [-10 + 10j, -1 - 3j]]
    )
    assert out.legs == [0, 1]
    np.testing.assert_allclose(out.data, ref, atol=1e-15)


def test_dimension_order():
    """contraction.rs:232-261: matrix axis 1 is the input index."""
    ket0 = OTensor([0], np.array([1 + 0j, 0 + 0j]))
    mat = OTensor([1, 0], np.array([[1, 2], [3, 4]], dtype=np.complex128))
    out = contract_network([ket0, mat], [(0, 1)])
    assert out.legs == [1]
    np.testing.assert_allclose(out.data, np.array([1, 3], dtype=np.complex128))


def test_gate_adjoint_consistency():
    """gates.rs:585-607: adjoint == conj-transpose; also unitarity."""
    rng = np.random.default_rng(42)
    params = {"u": 3, "rx": 1, "ry": 1, "rz": 1, "cp": 1, "fsim": 2}
    for name in gate_names():
        angles = rng.uniform(-np.pi, np.pi, params.get(name, 0)).tolist()
        g = load_gate(name, angles)
        adj = load_gate_adjoint(name, angles)
        np.testing.assert_allclose(adj, matrix_adjoint(g), atol=1e-15)
        # all 18 gates are unitary: G @ G^dagger == I
        n = int(np.sqrt(g.size))
        gm = g.reshape(n, n)
        np.testing.assert_allclose(
            gm @ adj.reshape(n, n), np.eye(n), atol=1e-14, err_msg=name
        )


def test_gate_exact_values():
    """Spot exact constants (gates.rs:150-556)."""
    s2 = 1 / np.sqrt(2)
    np.testing.assert_array_equal(load_gate("x"), [[0, 1], [1, 0]])
    np.testing.assert_allclose(load_gate("h"), [[s2, s2], [s2, -s2]])
    sy = load_gate("sy")
    np.testing.assert_array_equal(
        sy, [[0.5 + 0.5j, -0.5 - 0.5j], [0.5 + 0.5j, 0.5 + 0.5j]]
    )
    fsim = load_gate("fsim", [0.3, 0.2]).reshape(4, 4)
    assert fsim[1, 1] == complex(np.cos(0.3), 0)
    assert fsim[1, 2] == complex(0, -np.sin(0.3))
    np.testing.assert_allclose(fsim[3, 3], np.exp(-0.2j))


def test_scalar_result_and_rank0():
    a = OTensor([0, 1], np.arange(4, dtype=np.complex128).reshape(2, 2))
    b = OTensor([0, 1], (1j * np.arange(4, dtype=np.complex128)).reshape(2, 2))
    out = contract_tensors(a, b)
    assert out.legs == []
    np.testing.assert_allclose(out.data, np.sum(a.data * b.data))
