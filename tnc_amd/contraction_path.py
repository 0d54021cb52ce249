"""Contraction paths, mirroring tnc/src/contractionpath.rs.

A ContractionPath has nested per-composite paths plus a toplevel pair list
(contractionpath.rs:30-35). SSA and replace-left conversions follow
contractionpath.rs:180-215 exactly.
"""

from __future__ import annotations

from typing import Dict, List, Tuple


class ContractionPath:
    """contractionpath.rs:30-35."""

    __slots__ = ("nested", "toplevel")

    def __init__(self, nested=None, toplevel=None):
        self.nested: Dict[int, ContractionPath] = dict(nested or {})
        self.toplevel: List[Tuple[int, int]] = [tuple(p) for p in (toplevel or [])]

    @classmethod
    def simple(cls, path):
        return cls(toplevel=path)

    @classmethod
    def single(cls, a, b):
        return cls(toplevel=[(a, b)])

    def __len__(self):
        return len(self.toplevel)

    def is_simple(self):
        return not self.nested

    def is_empty(self):
        return not self.toplevel and not self.nested

    def into_simple(self):
        """The toplevel replace path; asserts no nested parts
        (contractionpath.rs into_simple semantics)."""
        assert self.is_simple(), "path has nested parts"
        return self.toplevel

    def __eq__(self, other):
        return (
            isinstance(other, ContractionPath)
            and self.nested == other.nested
            and self.toplevel == other.toplevel
        )

    def __repr__(self):
        return f"ContractionPath(nested={self.nested}, toplevel={self.toplevel})"


def path(*toplevel, nested=None):
    """Convenience mirror of the path! macro (contractionpath.rs:154-167)."""
    return ContractionPath(
        nested={k: v if isinstance(v, ContractionPath) else path(*v) for k, v in (nested or {}).items()},
        toplevel=list(toplevel),
    )


def ssa_ordering(raw_path, n):
    """contractionpath.rs:180-192: normalize (u1,u2,u3) triplets to strict SSA."""
    ssa_path = []
    hs = {}
    path_len = n
    for u1, u2, u3 in raw_path:
        t1 = hs[u1] if u1 >= path_len else u1
        t2 = hs[u2] if u2 >= path_len else u2
        hs.setdefault(u3, n)
        n += 1
        ssa_path.append((t1, t2))
    return ContractionPath.simple(ssa_path)


def ssa_replace_ordering(p: ContractionPath) -> ContractionPath:
    """SSA -> replace-left (contractionpath.rs:197-215).

    Contraction i's SSA result id is len(path)+1+i; it replaces the slot of
    its (resolved) left input.
    """
    nested = {i: ssa_replace_ordering(lp) for i, lp in p.nested.items()}
    hs = {}
    toplevel = []
    n = len(p.toplevel) + 1
    for t0, t1 in p.toplevel:
        new_t0 = hs.get(t0, t0)
        new_t1 = hs.get(t1, t1)
        assert n not in hs
        hs[n] = new_t0
        n += 1
        toplevel.append((new_t0, new_t1))
    return ContractionPath(nested=nested, toplevel=toplevel)


def validate_path(p: ContractionPath):
    """paths.rs:87-100: no contracting of already-consumed tensors."""
    contracted = []
    for nested in p.nested.values():
        validate_path(nested)
    for u, v in p.toplevel:
        assert u not in contracted, f"Contracting already contracted tensors: {u}, path: {p}"
        contracted.append(v)


def flatten_network(tn, replace_path: ContractionPath):
    """Flatten a (possibly nested) network + replace-left path into a flat
    leaf list and a flat (i, j) step list over global leaf indices.

    Equivalent to the recursive walk of contraction.rs:35-68: a composite
    child's nested path runs first (on global indices of its leaves); the
    composite's result slot is the global index holding its final tensor.
    Used to drive the device executor with a single linear plan.
    """
    from .tensor import CompositeTensor

    leaves = []
    steps = []

    def visit(node, p: ContractionPath):
        # returns (slot_map: local index -> global index of final tensor)
        assert isinstance(node, CompositeTensor)
        slot = []
        for idx, child in enumerate(node.tensors):
            if isinstance(child, CompositeTensor):
                inner = p.nested.get(idx)
                assert inner is not None, f"composite child {idx} without nested path"
                slot.append(visit(child, inner))
            else:
                slot.append(len(leaves))
                leaves.append(child)
        for i, j in p.toplevel:
            steps.append((slot[i], slot[j]))
            # result replaces slot i (replace-left)
        # final tensor of this composite: the slot that remains
        if p.toplevel:
            consumed = {slot[j] for _, j in p.toplevel}
            # the first step's left slot chain always ends at the final; find
            # the single non-consumed slot among those participating
            remaining = [g for g in slot if g not in consumed]
            assert len(remaining) == 1, "path does not fully contract composite"
            return remaining[0]
        assert len(slot) == 1
        return slot[0]

    final = visit(tn, replace_path)
    return leaves, steps, final
