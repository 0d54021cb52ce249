"""Host-side tensor data model, mirroring tnc/src/tensornetwork/tensor.rs and
tensordata.rs.

A LeafTensor is legs (edge ids) + bond dims + TensorData; a CompositeTensor
is a list of child tensors (tensor.rs:44-63). Leg set algebra reproduces the
reference's ordering contract exactly (tensor.rs:629-725): it defines the
output layout of every contraction and must match bit-for-bit.

Data lives host-side as numpy complex128 until the executor uploads it; on
the GPU the executor owns device buffers (see csrc/).
"""

from __future__ import annotations

import math
from typing import Iterable, Optional

import numpy as np

from . import gates as _gates


class TensorData:
    """Mirror of TensorData (tensordata.rs:17-27): None | Gate | Matrix.

    (File/HDF5 variant is out of scope, DESIGN.md.)
    """

    __slots__ = ("kind", "gate", "angles", "adjoint_flag", "matrix")

    NONE = "none"
    GATE = "gate"
    MATRIX = "matrix"

    def __init__(self, kind, gate=None, angles=None, adjoint_flag=False, matrix=None):
        self.kind = kind
        self.gate = gate
        self.angles = list(angles) if angles else []
        self.adjoint_flag = adjoint_flag
        self.matrix = matrix

    @classmethod
    def none(cls):
        return cls(cls.NONE)

    @classmethod
    def from_gate(cls, name: str, angles=(), adjoint: bool = False):
        return cls(cls.GATE, gate=name, angles=angles, adjoint_flag=adjoint)

    @classmethod
    def new_from_data(cls, dimensions, data) -> "TensorData":
        """tensordata.rs:32-34."""
        arr = np.asarray(data, dtype=np.complex128).reshape(tuple(dimensions))
        return cls(cls.MATRIX, matrix=arr)

    def into_data(self) -> np.ndarray:
        """Materialize (tensordata.rs:37-56). Gate builds are host-side (K2)."""
        if self.kind == self.MATRIX:
            return self.matrix
        if self.kind == self.GATE:
            if self.adjoint_flag:
                return _gates.load_gate_adjoint(self.gate, self.angles)
            return _gates.load_gate(self.gate, self.angles)
        raise ValueError("Cannot convert uncontracted tensor to data")

    def adjoint(self) -> "TensorData":
        """tensordata.rs:59-69."""
        if self.kind == self.NONE:
            return TensorData.none()
        if self.kind == self.GATE:
            return TensorData.from_gate(self.gate, self.angles, not self.adjoint_flag)
        return TensorData(self.MATRIX, matrix=_gates.matrix_adjoint(self.matrix))

    def __repr__(self):
        if self.kind == self.GATE:
            return f"TensorData.Gate({self.gate!r}, {self.angles}, adj={self.adjoint_flag})"
        if self.kind == self.MATRIX:
            return f"TensorData.Matrix(shape={self.matrix.shape})"
        return "TensorData.None"


class LeafTensor:
    """Mirror of LeafTensor (tensor.rs:437-496)."""

    __slots__ = ("_legs", "_bond_dims", "tensordata")

    def __init__(self, legs, bond_dims, data: Optional[TensorData] = None):
        legs = list(legs)
        bond_dims = list(bond_dims)
        assert len(legs) == len(bond_dims)
        self._legs = legs
        self._bond_dims = bond_dims
        self.tensordata = data if data is not None else TensorData.none()

    # --- constructors (tensor.rs:476-496) ---
    @classmethod
    def new_from_map(cls, legs, bond_dims_map):
        return cls(legs, [bond_dims_map[l] for l in legs])

    @classmethod
    def new_from_const(cls, legs, bond_dim):
        return cls(legs, [bond_dim] * len(legs))

    # --- accessors ---
    @property
    def legs(self):
        return self._legs

    @property
    def bond_dims(self):
        return self._bond_dims

    @property
    def shape(self):
        return tuple(self._bond_dims)

    def dims(self) -> int:
        return len(self._legs)

    def size(self) -> float:
        """Number of elements as float (tensor.rs:571-573)."""
        return float(math.prod(self._bond_dims)) if self._bond_dims else 1.0

    def set_tensor_data(self, data: TensorData):
        self.tensordata = data

    def shallow_clone(self) -> "LeafTensor":
        return LeafTensor(self._legs, self._bond_dims)

    def is_leaf(self):
        return True

    def is_composite(self):
        return False

    # --- leg set algebra (tensor.rs:629-725) ---
    def difference(self, other: "LeafTensor") -> "LeafTensor":
        so = set(other._legs)
        pairs = [(l, d) for l, d in zip(self._legs, self._bond_dims) if l not in so]
        return LeafTensor([p[0] for p in pairs], [p[1] for p in pairs])

    def union(self, other: "LeafTensor") -> "LeafTensor":
        ss = set(self._legs)
        legs = list(self._legs)
        dims = list(self._bond_dims)
        for l, d in zip(other._legs, other._bond_dims):
            if l not in ss:
                legs.append(l)
                dims.append(d)
        return LeafTensor(legs, dims)

    def intersection(self, other: "LeafTensor") -> "LeafTensor":
        so = set(other._legs)
        pairs = [(l, d) for l, d in zip(self._legs, self._bond_dims) if l in so]
        return LeafTensor([p[0] for p in pairs], [p[1] for p in pairs])

    def symmetric_difference(self, other: "LeafTensor") -> "LeafTensor":
        """A-only legs in A order, then B-only in B order (tensor.rs:709-725)."""
        a = self.difference(other)
        b = other.difference(self)
        return LeafTensor(a._legs + b._legs, a._bond_dims + b._bond_dims)

    def __sub__(self, other):
        return self.difference(other)

    def __or__(self, other):
        return self.union(other)

    def __and__(self, other):
        return self.intersection(other)

    def __xor__(self, other):
        return self.symmetric_difference(other)

    def __repr__(self):
        return f"LeafTensor(legs={self._legs}, dims={self._bond_dims}, {self.tensordata!r})"


class CompositeTensor:
    """Mirror of CompositeTensor (tensor.rs:189-403)."""

    __slots__ = ("_tensors",)

    def __init__(self, tensors: Iterable = ()):  # children: Leaf or Composite
        self._tensors = list(tensors)

    @property
    def tensors(self):
        return self._tensors

    def tensor(self, i):
        return self._tensors[i]

    def __len__(self):
        return len(self._tensors)

    def is_leaf(self):
        return False

    def is_composite(self):
        return True

    def push_tensor(self, t):
        self._tensors.append(t)

    def push_tensors(self, ts):
        self._tensors.extend(ts)

    def total_num_tensors(self) -> int:
        return sum(
            t.total_num_tensors() if isinstance(t, CompositeTensor) else 1
            for t in self._tensors
        )

    def is_connected(self) -> bool:
        """All children pairwise-connected through shared legs
        (tensor.rs:368-389; children must be leaves)."""
        from .utils import UnionFind

        n = len(self._tensors)
        uf = UnionFind(n)
        for i in range(n):
            for j in range(i + 1, n):
                t1, t2 = self._tensors[i], self._tensors[j]
                assert not isinstance(t1, CompositeTensor)
                assert not isinstance(t2, CompositeTensor)
                if (t1 & t2).legs:
                    uf.union(i, j)
        return uf.count_sets() == 1

    def external_tensor(self) -> LeafTensor:
        """Open legs after full contraction (tensor.rs:392-403): fold of
        symmetric differences over children, in order."""
        acc = LeafTensor([], [])
        for t in self._tensors:
            leaf = t.external_tensor() if isinstance(t, CompositeTensor) else t
            acc = acc ^ leaf
        return acc

    def __repr__(self):
        return f"CompositeTensor({len(self._tensors)} children)"
