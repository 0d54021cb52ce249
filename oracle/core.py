"""Reference-semantics restatement of TNC's contraction executor (numpy).

Restates, exactly:
- leg set algebra / output-leg ordering: tnc/src/tensornetwork/tensor.rs:629-725
- pairwise contraction:                  tnc/src/tensornetwork/contraction.rs:70-116
- replace-left network walk:             tnc/src/tensornetwork/contraction.rs:35-68

Tensors are row-major complex128 ndarrays (ArrayD<Complex64>,
tnc/src/tensornetwork/tensordata.rs:13). Labels are integer edge ids.
"""

from __future__ import annotations

import numpy as np


class OTensor:
    """A leaf tensor: integer legs + row-major complex128 data.

    Mirrors LeafTensor (tnc/src/tensornetwork/tensor.rs:437-496): legs and
    bond_dims have the same length and order; data shape == bond_dims.
    """

    __slots__ = ("legs", "data")

    def __init__(self, legs, data):
        legs = list(legs)
        data = np.asarray(data)
        assert data.ndim == len(legs), (legs, data.shape)
        self.legs = legs
        self.data = data

    @property
    def shape(self):
        return self.data.shape


def _diff_order(legs_a, dims_a, legs_b):
    """Legs of A not in B, in A's order (tensor.rs:629-639)."""
    sb = set(legs_b)
    out_legs, out_dims = [], []
    for leg, dim in zip(legs_a, dims_a):
        if leg not in sb:
            out_legs.append(leg)
            out_dims.append(dim)
    return out_legs, out_dims


def symmetric_difference(legs_a, dims_a, legs_b, dims_b):
    """A-only legs in A order, then B-only legs in B order (tensor.rs:709-725)."""
    la, da = _diff_order(legs_a, dims_a, legs_b)
    lb, db = _diff_order(legs_b, dims_b, legs_a)
    return la + lb, da + db


def intersection(legs_a, dims_a, legs_b):
    """Legs of A also in B, in A order (tensor.rs:683-693)."""
    sb = set(legs_b)
    out_legs, out_dims = [], []
    for leg, dim in zip(legs_a, dims_a):
        if leg in sb:
            out_legs.append(leg)
            out_dims.append(dim)
    return out_legs, out_dims


def union(legs_a, dims_a, legs_b, dims_b):
    """All of A's legs in A order, then B-only legs in B order (tensor.rs:655-667)."""
    sa = set(legs_a)
    out_legs, out_dims = list(legs_a), list(dims_a)
    for leg, dim in zip(legs_b, dims_b):
        if leg not in sa:
            out_legs.append(leg)
            out_dims.append(dim)
    return out_legs, out_dims


def contract_ndarrays(out_labels, a_labels, a_data, b_labels, b_data, optimize=True):
    """out[out_labels] = sum over shared labels of A*B (contraction.rs:88-116).

    Direct numpy restatement of the tblis::tensor_mult call
    (contraction.rs:111-113): labels present in both inputs are contracted;
    the caller fixes the output label order. K may be 1 (outer product); the
    output may be rank 0 (scalar).
    """
    # np.einsum's integer-label interface needs small nonneg ints; remap.
    remap = {}
    for lab in list(a_labels) + list(b_labels):
        if lab not in remap:
            remap[lab] = len(remap)
    a_l = [remap[l] for l in a_labels]
    b_l = [remap[l] for l in b_labels]
    o_l = [remap[l] for l in out_labels]
    return np.einsum(a_data, a_l, b_data, b_l, o_l, optimize=optimize)


def contract_tensors(ta: OTensor, tb: OTensor) -> OTensor:
    """One path step (contraction.rs:70-86): out legs = ta ^ tb."""
    out_legs, _ = symmetric_difference(ta.legs, ta.shape, tb.legs, tb.shape)
    out = contract_ndarrays(out_legs, ta.legs, ta.data, tb.legs, tb.data)
    return OTensor(out_legs, out)


def contract_network(tensors, path) -> OTensor:
    """Replace-left walk (contraction.rs:35-68).

    `tensors` is a flat list of OTensor (or nested lists for composite
    children); `path` is a dict-free structure: either a list of (i, j)
    pairs (simple path) or an object with .nested (dict idx->path) and
    .toplevel. Nested composites are contracted first (contraction.rs:42-49),
    then the toplevel replace-left loop runs (contraction.rs:52-57).
    """
    slots = list(tensors)
    nested = getattr(path, "nested", None)
    toplevel = getattr(path, "toplevel", path)
    if nested:
        for idx, inner in nested.items():
            slots[idx] = contract_network(slots[idx], inner)
    for i, j in toplevel:
        ti = slots[i]
        tj = slots[j]
        assert ti is not None and tj is not None, (i, j)
        slots[i] = contract_tensors(ti, tj)
        slots[j] = None
    remaining = [t for t in slots if t is not None]
    assert len(remaining) == 1, f"path left {len(remaining)} tensors"
    return remaining[0]
