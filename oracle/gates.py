"""Gate matrices with the exact constants of tnc/src/gates.rs:150-556.

Two-qubit gates are shaped (2,2,2,2) exactly like the reference
(into_shape_with_order((2,2,2,2)), e.g. gates.rs:423-426). The adjoint of a
matrix-like tensor swaps the first half of the axes with the second half and
conjugates (gates.rs:83-101).
"""

from __future__ import annotations

import math

import numpy as np

_S2 = 1.0 / math.sqrt(2.0)


def matrix_adjoint(data: np.ndarray) -> np.ndarray:
    """gates.rs:83-101: permute axes (half..n, 0..half), then conjugate."""
    if data.ndim > 0:
        n = data.ndim
        assert (n & (n - 1)) == 0, "ndim must be a power of two"
        half = n // 2
        perm = list(range(half, n)) + list(range(half))
        data = np.transpose(data, perm)
    return np.conj(data)


def _g_x(angles):
    assert not angles
    return np.array([[0, 1], [1, 0]], dtype=np.complex128)


def _g_y(angles):
    assert not angles
    return np.array([[0, -1j], [1j, 0]], dtype=np.complex128)


def _g_z(angles):
    assert not angles
    return np.array([[1, 0], [0, -1]], dtype=np.complex128)


def _g_h(angles):
    assert not angles
    return np.array([[_S2, _S2], [_S2, -_S2]], dtype=np.complex128)


def _g_t(angles):
    assert not angles
    return np.array([[1, 0], [0, complex(_S2, _S2)]], dtype=np.complex128)


def _g_u(angles):
    # OpenQASM 3 U gate (gates.rs:252-287)
    theta, phi, lam = angles
    sin, cos = math.sin(theta / 2.0), math.cos(theta / 2.0)
    return np.array(
        [
            [complex(cos, 0.0), -np.exp(1j * lam) * sin],
            [np.exp(1j * phi) * sin, np.exp(1j * (phi + lam)) * cos],
        ],
        dtype=np.complex128,
    )


def _g_sx(angles):
    assert not angles
    a, b = complex(0.5, 0.5), complex(0.5, -0.5)
    return np.array([[a, b], [b, a]], dtype=np.complex128)


def _g_sy(angles):
    # NOTE: reproduced exactly as written in gates.rs:318-323 ([[a,b],[a,a]]).
    assert not angles
    a, b = complex(0.5, 0.5), complex(-0.5, -0.5)
    return np.array([[a, b], [a, a]], dtype=np.complex128)


def _g_sz(angles):
    assert not angles
    return np.array([[1, 0], [0, 1j]], dtype=np.complex128)


def _g_rx(angles):
    (theta,) = angles
    sin, cos = math.sin(theta / 2.0), math.cos(theta / 2.0)
    return np.array([[cos, -1j * sin], [-1j * sin, cos]], dtype=np.complex128)


def _g_ry(angles):
    (theta,) = angles
    sin, cos = math.sin(theta / 2.0), math.cos(theta / 2.0)
    return np.array([[cos, -sin], [sin, cos]], dtype=np.complex128)


def _g_rz(angles):
    (theta,) = angles
    return np.array(
        [[np.exp(-0.5j * theta), 0], [0, np.exp(0.5j * theta)]], dtype=np.complex128
    )


def _mat4(rows):
    return np.array(rows, dtype=np.complex128).reshape(2, 2, 2, 2)


def _g_cx(angles):
    assert not angles
    return _mat4([[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 0, 1], [0, 0, 1, 0]])


def _g_cz(angles):
    assert not angles
    return _mat4([[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 1, 0], [0, 0, 0, -1]])


def _g_swap(angles):
    assert not angles
    return _mat4([[1, 0, 0, 0], [0, 0, 1, 0], [0, 1, 0, 0], [0, 0, 0, 1]])


def _g_cp(angles):
    (theta,) = angles
    e = np.exp(1j * theta)
    return _mat4([[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 1, 0], [0, 0, 0, e]])


def _g_iswap(angles):
    assert not angles
    return _mat4([[1, 0, 0, 0], [0, 0, 1j, 0], [0, 1j, 0, 0], [0, 0, 0, 1]])


def _g_fsim(angles):
    # gates.rs:532-555 (cirq FSimGate convention)
    theta, phi = angles
    a = complex(math.cos(theta), 0.0)
    b = complex(0.0, -math.sin(theta))
    c = np.exp(complex(0.0, -phi))
    return _mat4([[1, 0, 0, 0], [0, a, b, 0], [0, b, a, 0], [0, 0, 0, c]])


_GATES = {
    "x": _g_x,
    "y": _g_y,
    "z": _g_z,
    "h": _g_h,
    "t": _g_t,
    "u": _g_u,
    "sx": _g_sx,
    "sy": _g_sy,
    "sz": _g_sz,
    "rx": _g_rx,
    "ry": _g_ry,
    "rz": _g_rz,
    "cx": _g_cx,
    "cz": _g_cz,
    "swap": _g_swap,
    "cp": _g_cp,
    "iswap": _g_iswap,
    "fsim": _g_fsim,
}


def load_gate(name: str, angles=()) -> np.ndarray:
    """gates.rs:51-57."""
    return _GATES[name](list(angles))


def load_gate_adjoint(name: str, angles=()) -> np.ndarray:
    """gates.rs:61-67 (generic conjugate-transpose; the reference's
    specialized adjoints are value-equal, pinned by gates.rs:585-607)."""
    return matrix_adjoint(load_gate(name, angles))


def gate_names():
    return sorted(_GATES)
