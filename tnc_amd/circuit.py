"""Quantum-circuit -> tensor-network builder, mirroring
tnc/src/builders/circuit_builder.rs.

Edge numbering contract (circuit_builder.rs:196-220): appending a k-qubit
gate creates k new edges first, then chains the qubits' previous open edges;
the gate tensor's legs are [new..., old...]. Gate matrices are indexed
[out, in] per qubit, so axis k is the output of qubit k and axis k+count the
input.
"""

from __future__ import annotations

from typing import List

import numpy as np

from .tensor import CompositeTensor, LeafTensor, TensorData


class Permutor:
    """Final-tensor axis permutation to natural qubit order
    (circuit_builder.rs:77-122)."""

    def __init__(self, target_leg_order: List[int]):
        self.target_leg_order = list(target_leg_order)

    def is_identity(self):
        return not self.target_leg_order

    @staticmethod
    def permutation_between(given, target):
        """The permutation perm with [given[i] for i in ...] — matching the
        reference's permutation crate semantics (circuit_builder.rs:117-121):
        applying the result to `given` yields `target`."""
        pos = {v: i for i, v in enumerate(given)}
        # perm maps target position -> given position
        return [pos[t] for t in target]

    def apply(self, legs, bond_dims, data: np.ndarray):
        """circuit_builder.rs:89-106. Returns (legs, bond_dims, data)."""
        if self.is_identity():
            return legs, bond_dims, data
        perm = self.permutation_between(legs, self.target_leg_order)
        new_legs = [legs[p] for p in perm]
        new_dims = [bond_dims[p] for p in perm]
        return new_legs, new_dims, np.ascontiguousarray(np.transpose(data, perm))

    def apply_leaf(self, tensor: LeafTensor) -> LeafTensor:
        if self.is_identity():
            return tensor
        legs, dims, data = self.apply(
            tensor.legs, tensor.bond_dims, tensor.tensordata.into_data()
        )
        out = LeafTensor(legs, dims)
        out.set_tensor_data(TensorData(TensorData.MATRIX, matrix=data))
        return out


class _Qubit:
    __slots__ = ("index",)

    def __init__(self, index):
        self.index = index


class QuantumRegister:
    """circuit_builder.rs:21-67."""

    def __init__(self, base, size):
        self.base = base
        self.size = size

    def qubit(self, index):
        assert index < self.size
        return _Qubit(self.base + index)

    def qubits(self):
        return [_Qubit(i) for i in range(self.base, self.base + self.size)]

    def __len__(self):
        return self.size


def _ket0():
    return TensorData.new_from_data([2], [1.0, 0.0])


def _ket1():
    return TensorData.new_from_data([2], [0.0, 1.0])


class Circuit:
    """circuit_builder.rs:124-327."""

    def __init__(self):
        self.open_edges: List[int] = []
        self.next_edge = 0
        self.tensor_network = CompositeTensor()

    def num_qubits(self):
        return len(self.open_edges)

    def qubit(self, index) -> _Qubit:
        """Absolute qubit handle across registers (importer convenience)."""
        assert 0 <= index < self.num_qubits()
        return _Qubit(index)

    def _new_edge(self):
        e = self.next_edge
        self.next_edge += 1
        return e

    def allocate_register(self, size) -> QuantumRegister:
        """circuit_builder.rs:176-194: qubits start in |0>."""
        base = self.num_qubits()
        for _ in range(size):
            e = self._new_edge()
            self.open_edges.append(e)
            t = LeafTensor.new_from_const([e], 2)
            t.set_tensor_data(_ket0())
            self.tensor_network.push_tensor(t)
        return QuantumRegister(base, size)

    def append_gate(self, gate: TensorData, qubits):
        """circuit_builder.rs:196-220: legs = new edges then old edges."""
        idx = [q.index for q in qubits]
        assert len(set(idx)) == len(idx), "Qubit arguments must be unique"
        old_edges = [self.open_edges[i] for i in idx]
        new_edges = [self.next_edge + k for k in range(len(idx))]
        edges = new_edges + old_edges
        self.next_edge += len(idx)
        for i, ne in zip(idx, new_edges):
            self.open_edges[i] = ne
        t = LeafTensor.new_from_const(edges, 2)
        t.set_tensor_data(gate)
        self.tensor_network.push_tensor(t)

    def into_amplitude_network(self, bitstring: str):
        """circuit_builder.rs:235-262. '*' leaves the qubit's edge open."""
        assert len(bitstring) == self.num_qubits()
        final_legs = []
        for c, e in zip(bitstring, self.open_edges):
            if c == "*":
                final_legs.append(e)
                continue
            if c == "0":
                bra = _ket0()
            elif c == "1":
                bra = _ket1()
            else:
                raise ValueError("Only 0, 1 and * are allowed in bitstring")
            t = LeafTensor.new_from_const([e], 2)
            t.set_tensor_data(bra)
            self.tensor_network.push_tensor(t)
        return self.tensor_network, Permutor(final_legs)

    def into_statevector_network(self):
        """circuit_builder.rs:270-273."""
        return self.into_amplitude_network("*" * self.num_qubits())

    @staticmethod
    def _tensor_adjoint(tensor: LeafTensor, leg_offset: int) -> LeafTensor:
        """circuit_builder.rs:278-297: transpose leg halves, offset, adjoint data."""
        half = len(tensor.legs) // 2
        legs = [l + leg_offset for l in tensor.legs[half:] + tensor.legs[:half]]
        dims = tensor.bond_dims[half:] + tensor.bond_dims[:half]
        out = LeafTensor(legs, dims)
        out.set_tensor_data(tensor.tensordata.adjoint())
        return out

    def into_expectation_value_network(self) -> CompositeTensor:
        """circuit_builder.rs:304-326: mirrored adjoint circuit + Z layer."""
        offset = self.next_edge
        adjoints = [
            self._tensor_adjoint(t, offset) for t in self.tensor_network.tensors
        ]
        self.tensor_network.push_tensors(adjoints)
        for e in self.open_edges:
            t = LeafTensor.new_from_const([e, e + offset], 2)
            t.set_tensor_data(TensorData.from_gate("z"))
            self.tensor_network.push_tensor(t)
        return self.tensor_network
