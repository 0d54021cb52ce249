"""Product gate registry, mirroring tnc/src/gates.rs (18 gates, runtime
extensible via register_gate, gates.rs:41-47).

Values are independently cross-checked against the oracle's restatement and
the reference's exact-value tests (see tests/). Two-qubit gates are shaped
(2,2,2,2) like the reference (gates.rs:423-426).
"""

from __future__ import annotations

import math

import numpy as np

_S2 = 1.0 / math.sqrt(2.0)


def matrix_adjoint(data: np.ndarray) -> np.ndarray:
    """Swap first/second half of axes and conjugate (gates.rs:83-101)."""
    if data.ndim > 0:
        n = data.ndim
        assert (n & (n - 1)) == 0
        half = n // 2
        data = np.transpose(data, list(range(half, n)) + list(range(half)))
    return np.conj(data)


def _m2(rows):
    return np.array(rows, dtype=np.complex128)


def _m4(rows):
    return np.array(rows, dtype=np.complex128).reshape(2, 2, 2, 2)


def _angles(angles, n, name):
    if len(angles) != n:
        raise ValueError(f"Expected {n} angles for {name}, but got {len(angles)}.")
    return angles


_REGISTRY = {}


def register_gate(name, fn, adjoint_fn=None):
    """gates.rs:41-47 (names must be lowercase)."""
    assert name == name.lower(), "Gate name must be lowercase."
    _REGISTRY[name] = (fn, adjoint_fn)


def load_gate(name, angles=()):
    """gates.rs:51-57."""
    if name not in _REGISTRY:
        raise KeyError(f"Gate '{name}' not found.")
    return _REGISTRY[name][0](list(angles))


def load_gate_adjoint(name, angles=()):
    """gates.rs:61-67."""
    if name not in _REGISTRY:
        raise KeyError(f"Gate '{name}' not found.")
    fn, adj = _REGISTRY[name]
    if adj is not None:
        return adj(list(angles))
    return matrix_adjoint(fn(list(angles)))


def is_gate_known(name):
    return name in _REGISTRY


# --- the 18 built-in gates (gates.rs:150-556) ---

register_gate("x", lambda a: (_angles(a, 0, "x"), _m2([[0, 1], [1, 0]]))[1])
register_gate("y", lambda a: (_angles(a, 0, "y"), _m2([[0, -1j], [1j, 0]]))[1])
register_gate("z", lambda a: (_angles(a, 0, "z"), _m2([[1, 0], [0, -1]]))[1])
register_gate("h", lambda a: (_angles(a, 0, "h"), _m2([[_S2, _S2], [_S2, -_S2]]))[1])
register_gate(
    "t", lambda a: (_angles(a, 0, "t"), _m2([[1, 0], [0, complex(_S2, _S2)]]))[1]
)


def _u(a):
    theta, phi, lam = _angles(a, 3, "u")
    s, c = math.sin(theta / 2), math.cos(theta / 2)
    return _m2(
        [
            [complex(c, 0.0), -np.exp(1j * lam) * s],
            [np.exp(1j * phi) * s, np.exp(1j * (phi + lam)) * c],
        ]
    )


register_gate("u", _u)

register_gate(
    "sx",
    lambda a: (
        _angles(a, 0, "sx"),
        _m2([[0.5 + 0.5j, 0.5 - 0.5j], [0.5 - 0.5j, 0.5 + 0.5j]]),
    )[1],
)
# sy reproduced exactly as the reference defines it (gates.rs:318-323)
register_gate(
    "sy",
    lambda a: (
        _angles(a, 0, "sy"),
        _m2([[0.5 + 0.5j, -0.5 - 0.5j], [0.5 + 0.5j, 0.5 + 0.5j]]),
    )[1],
)
register_gate("sz", lambda a: (_angles(a, 0, "sz"), _m2([[1, 0], [0, 1j]]))[1])


def _rx(a):
    (theta,) = _angles(a, 1, "rx")
    s, c = math.sin(theta / 2), math.cos(theta / 2)
    return _m2([[c, -1j * s], [-1j * s, c]])


def _ry(a):
    (theta,) = _angles(a, 1, "ry")
    s, c = math.sin(theta / 2), math.cos(theta / 2)
    return _m2([[c, -s], [s, c]])


def _rz(a):
    (theta,) = _angles(a, 1, "rz")
    return _m2([[np.exp(-0.5j * theta), 0], [0, np.exp(0.5j * theta)]])


register_gate("rx", _rx)
register_gate("ry", _ry)
register_gate("rz", _rz)

register_gate(
    "cx",
    lambda a: (
        _angles(a, 0, "cx"),
        _m4([[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 0, 1], [0, 0, 1, 0]]),
    )[1],
)
register_gate(
    "cz",
    lambda a: (
        _angles(a, 0, "cz"),
        _m4([[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 1, 0], [0, 0, 0, -1]]),
    )[1],
)
register_gate(
    "swap",
    lambda a: (
        _angles(a, 0, "swap"),
        _m4([[1, 0, 0, 0], [0, 0, 1, 0], [0, 1, 0, 0], [0, 0, 0, 1]]),
    )[1],
)


def _cp(a):
    (theta,) = _angles(a, 1, "cp")
    return _m4(
        [[1, 0, 0, 0], [0, 1, 0, 0], [0, 0, 1, 0], [0, 0, 0, np.exp(1j * theta)]]
    )


register_gate("cp", _cp)
register_gate(
    "iswap",
    lambda a: (
        _angles(a, 0, "iswap"),
        _m4([[1, 0, 0, 0], [0, 0, 1j, 0], [0, 1j, 0, 0], [0, 0, 0, 1]]),
    )[1],
)


def _fsim(a):
    theta, phi = _angles(a, 2, "fsim")
    aa = complex(math.cos(theta), 0.0)
    bb = complex(0.0, -math.sin(theta))
    cc = np.exp(complex(0.0, -phi))
    return _m4([[1, 0, 0, 0], [0, aa, bb, 0], [0, bb, aa, 0], [0, 0, 0, cc]])


register_gate("fsim", _fsim)
