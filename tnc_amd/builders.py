"""Random and Sycamore circuit generators, mirroring
tnc/src/builders/random_circuit.rs and sycamore_circuit.rs.

RNG note (flagged deviation): the reference uses Rust StdRng;
exact stream reproduction is not required — benchmark networks are generated
once by these seeded builders (numpy PCG64) and frozen as fixtures that both
the oracle and the GPU path read.
"""

from __future__ import annotations

import numpy as np

from .circuit import Circuit
from .connectivity import (
    SYCAMORE_A,
    SYCAMORE_B,
    SYCAMORE_C,
    SYCAMORE_D,
    connectivity_edges,
)
from .tensor import CompositeTensor, TensorData


def _rng(seed_or_rng):
    if isinstance(seed_or_rng, np.random.Generator):
        return seed_or_rng
    return np.random.Generator(np.random.PCG64(seed_or_rng))


def random_circuit(
    qubits, rounds, single_qubit_probability, two_qubit_probability, rng, connectivity
) -> CompositeTensor:
    """random_circuit.rs:29-80: per round, Bernoulli sx/sy/sz on each qubit,
    then Bernoulli fsim(0.3, 0.2) on each connectivity pair (u,v < qubits);
    closed as a single-amplitude <0...0| network."""
    rng = _rng(rng)
    single_gates = ["sx", "sy", "sz"]
    edges = [(u, v) for (u, v) in connectivity_edges(connectivity) if u < qubits and v < qubits]

    circuit = Circuit()
    qr = circuit.allocate_register(qubits)
    for _ in range(1, rounds):
        for i in range(qubits):
            if rng.random() < single_qubit_probability:
                g = single_gates[rng.integers(0, 3)]
                circuit.append_gate(TensorData.from_gate(g), [qr.qubit(i)])
        for i, j in edges:
            if rng.random() < two_qubit_probability:
                circuit.append_gate(
                    TensorData.from_gate("fsim", [0.3, 0.2]), [qr.qubit(i), qr.qubit(j)]
                )
    return circuit.into_amplitude_network("0" * qubits)[0]


def sycamore_circuit(qubits, depth, rng) -> Circuit:
    """sycamore_circuit.rs:22-72: rounds of random sx/sy/sz on every qubit +
    fsim(pi/2, pi/6) layers cycling A,B,C,D,C,D,A,B; a final single-qubit
    layer closes the circuit. Qubit ids in the layer tables are 1-based."""
    assert qubits <= 49, "only Sycamore-sized circuits are supported"
    rng = _rng(rng)
    layer_cycle = [
        SYCAMORE_A, SYCAMORE_B, SYCAMORE_C, SYCAMORE_D,
        SYCAMORE_C, SYCAMORE_D, SYCAMORE_A, SYCAMORE_B,
    ]
    single_gates = ["sx", "sy", "sz"]
    circuit = Circuit()
    qr = circuit.allocate_register(qubits)
    for rnd in range(depth + 1):
        for i in range(qubits):
            g = single_gates[rng.integers(0, 3)]
            circuit.append_gate(TensorData.from_gate(g), [qr.qubit(i)])
        if rnd < depth:
            layer = layer_cycle[rnd % 8]
            for i, j in layer:
                if i > qubits or j > qubits:
                    continue
                circuit.append_gate(
                    TensorData.from_gate("fsim", [np.pi / 2, np.pi / 6]),
                    [qr.qubit(i - 1), qr.qubit(j - 1)],
                )
    return circuit
