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
from .tensor import CompositeTensor, LeafTensor, TensorData


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


def random_circuit_with_observable(
    qubits, rounds, single_qubit_probability, two_qubit_probability,
    observable_probability, rng, connectivity
) -> CompositeTensor:
    """random_circuit.rs:88-113: random observable placement, then the
    set-observable construction."""
    rng = _rng(rng)
    locations = [i for i in range(qubits) if rng.random() < observable_probability]
    return random_circuit_with_set_observable(
        qubits, rounds, single_qubit_probability, two_qubit_probability,
        locations, rng, connectivity,
    )


def random_circuit_with_set_observable(
    qubits, rounds, single_qubit_probability, two_qubit_probability,
    observable_location, rng, connectivity
) -> CompositeTensor:
    """random_circuit.rs:120-275: a sandwich network <psi| U' O U |psi> built
    directly — per qubit a (left, right) open-edge pair, observables first,
    then mirrored fsim/single-qubit layers on qubits that affect the
    observable, then a random shared initial state on each side.

    Edge numbering reproduces the reference exactly (pinned by the
    leg-layout tests at random_circuit.rs:287-418)."""
    from .tensor import LeafTensor
    from .tensorgeneration import random_sparse_tensor_data

    rng = _rng(rng)
    single_gates = [("sx", "sx"), ("sy", "sx"), ("sz", "sx")]  # (fwd, adj) per :132-145
    observables = ["x", "y", "z"]

    tn = CompositeTensor()
    open_edges = {}
    next_edge = 0

    final_state = []
    for i in range(qubits):
        if i in observable_location:
            open_edges[i] = (next_edge, next_edge + 1)
            next_edge += 2
            obs = observables[int(rng.integers(0, 3))]
            t = LeafTensor.new_from_const([open_edges[i][0], open_edges[i][1]], 2)
            t.set_tensor_data(TensorData.from_gate(obs))
            final_state.append(t)
        else:
            open_edges[i] = (0, 0)
    tn.push_tensors(final_state)

    edges = [(u, v) for (u, v) in connectivity_edges(connectivity)
             if u < qubits and v < qubits]
    gates = []
    for _ in range(1, rounds):
        for i, j in edges:
            if (rng.random() < two_qubit_probability
                    and (open_edges[i][0] != open_edges[i][1]
                         or open_edges[j][0] != open_edges[j][1])):
                if open_edges[i][0] != open_edges[i][1]:
                    li, ri = open_edges[i]
                else:
                    li = ri = next_edge
                    next_edge += 1
                if open_edges[j][0] != open_edges[j][1]:
                    lj, rj = open_edges[j]
                else:
                    lj = rj = next_edge
                    next_edge += 1
                left = LeafTensor.new_from_const(
                    [next_edge, next_edge + 1, li, lj], 2)
                left.set_tensor_data(TensorData.from_gate("fsim", [0.3, 0.2]))
                gates.append(left)
                right = LeafTensor.new_from_const(
                    [ri, rj, next_edge + 2, next_edge + 3], 2)
                right.set_tensor_data(
                    TensorData.from_gate("fsim", [0.3, 0.2], adjoint=True))
                gates.append(right)
                open_edges[i] = (next_edge, next_edge + 2)
                open_edges[j] = (next_edge + 1, next_edge + 3)
                next_edge += 4
        for i in range(qubits):
            li, ri = open_edges[i]
            if rng.random() < single_qubit_probability and li != ri:
                fwd, adj = single_gates[int(rng.integers(0, 3))]
                left = LeafTensor.new_from_const([next_edge, li], 2)
                left.set_tensor_data(TensorData.from_gate(fwd))
                gates.append(left)
                right = LeafTensor.new_from_const([ri, next_edge + 1], 2)
                right.set_tensor_data(TensorData.from_gate(adj, adjoint=True))
                gates.append(right)
                open_edges[i] = (next_edge, next_edge + 1)
                next_edge += 2
    tn.push_tensors(gates)

    initial = []
    for i in range(qubits):
        li, ri = open_edges[i]
        if li != ri:
            state = random_sparse_tensor_data([2], 1.0, rng)
            left = LeafTensor.new_from_const([li], 2)
            left.set_tensor_data(state)
            right = LeafTensor.new_from_const([ri], 2)
            right.set_tensor_data(state)
            initial.extend([left, right])
    tn.push_tensors(initial)
    return tn


def peps(length, depth, physical_dim, virtual_dim, layers):
    """PEPS sandwich structure <bra| PEPO^layers |ket> on a length x depth
    lattice (builders/peps.rs:446-467). Leaves carry legs/dims only (no
    data), like the reference. Edge numbering per layer block of
    total = p + vv + vh edges (p = L*D physical, vv = (L-1)*D row-internal
    virtual, vh = (D-1)*L column virtual; peps.rs doc comment): site (i, j)
    legs are [physical...] then row bonds (j-1, j) then column bonds
    (i-1, i), absent neighbors skipped.

    `layers` = 0 gives the inner product of two states."""
    assert length > 1, "PEPS should have length greater than 1"
    assert depth > 1, "PEPS should have depth greater than 1"
    L, D = length, depth
    p = L * D
    vv = (L - 1) * D
    vh = (D - 1) * L
    total = p + vv + vh

    def site_legs(i, j, phys, vbase):
        """phys: list of physical legs; vbase: base of this layer's virtual
        block (row bonds at vbase, column bonds at vbase + vv)."""
        legs = list(phys)
        dims = [physical_dim] * len(phys)
        if j > 0:
            legs.append(vbase + i * (L - 1) + j - 1)
            dims.append(virtual_dim)
        if j < L - 1:
            legs.append(vbase + i * (L - 1) + j)
            dims.append(virtual_dim)
        if i > 0:
            legs.append(vbase + vv + (i - 1) * L + j)
            dims.append(virtual_dim)
        if i < D - 1:
            legs.append(vbase + vv + i * L + j)
            dims.append(virtual_dim)
        return LeafTensor(legs, dims)

    tensors = []
    # |ket> layer: physical legs [0, p), virtuals at [p, total)
    for i in range(D):
        for j in range(L):
            tensors.append(site_legs(i, j, [i * L + j], p))
    # PEPO layers k: physical prev = k*total + index, next = (k+1)*total +
    # index, virtuals at (k+1)*total + p
    for k in range(layers):
        last, start = k * total, (k + 1) * total
        for i in range(D):
            for j in range(L):
                idx = i * L + j
                tensors.append(
                    site_legs(i, j, [last + idx, start + idx], start + p))
    # <bra| layer: physical = layers*total + index, virtuals directly at
    # (layers+1)*total (the final block has no further physical edges)
    last, start = layers * total, (layers + 1) * total
    for i in range(D):
        for j in range(L):
            tensors.append(site_legs(i, j, [last + i * L + j], start))
    return CompositeTensor(tensors)
