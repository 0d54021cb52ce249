"""Simulated-annealing contraction-TREE refinement (the quality tier on top
of PartitionSearch, standing in for the reference's cotengra HyperOptimizer
bridge, paths/hyperoptimization.rs:1-60 — which needs the cotengra Python
package and cannot be installed here).

The search state is the binary contraction tree itself; moves are
nearest-neighbour interchanges (NNI): for an internal node X = (P, B) with
P = (C, D), the alternatives X = ((C, B), D) and X = ((D, B), C) contract
the same leaf set (leg symmetric difference is associative/commutative, so
every node above X keeps its view — TNC's contract semantics,
tensor.rs:629-725). Only P's intermediate changes, so a move is evaluated
in O(1): delta = op(C,B or D,B) - op(C,D), with op the reference's
prod-of-union-dims count (contraction_cost.rs:49-52). Acceptance follows
the reference SA's log2-ratio rule with a log-interpolated temperature
schedule (simulated_annealing.rs:85-166) applied per move.

A size cap (peak out+a+b elements per step, contract_size_tensors
semantics, contraction_cost.rs:69-72) is enforced on acceptance: moves
whose new step peak exceeds the cap are rejected.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import numpy as np

from .contraction_path import ContractionPath
from .paths import BasicContractionPathResult, PartitionSearch, _replace_to_ssa
from .tensor import CompositeTensor, LeafTensor


class _View:
    """Leg->dim map with cached size (leg ORDER does not matter for cost)."""

    __slots__ = ("dims", "size")

    def __init__(self, dims: Dict[int, int], size: float):
        self.dims = dims
        self.size = size

    @classmethod
    def of(cls, legs, bond_dims):
        dims = dict(zip(legs, bond_dims))
        size = 1.0
        for d in dims.values():
            size *= d
        return cls(dims, size)

    def xor(self, other: "_View") -> "_View":
        dims = dict(self.dims)
        size = self.size
        for leg, d in other.dims.items():
            if leg in dims:
                size /= dims.pop(leg)
            else:
                dims[leg] = d
                size *= d
        return _View(dims, size)

    def union_size(self, other: "_View") -> float:
        shared = 1.0
        a, b = (self.dims, other.dims) if len(self.dims) <= len(other.dims) \
            else (other.dims, self.dims)
        for leg, d in a.items():
            if leg in b:
                shared *= d
        return self.size * other.size / shared


class _Tree:
    """Mutable binary contraction tree over n leaves.

    Nodes 0..n-1 are leaves; internal nodes are n..2n-2. Arrays index by
    node id. Internal node views/op costs are maintained incrementally.
    """

    def __init__(self, leaves: List[LeafTensor], toplevel):
        n = len(leaves)
        self.n = n
        total = 2 * n - 1
        self.left = [-1] * total
        self.right = [-1] * total
        self.parent = [-1] * total
        self.view: List[Optional[_View]] = [None] * total
        for i, t in enumerate(leaves):
            self.view[i] = _View.of(t.legs, t.bond_dims)
        self.op = [0.0] * total        # op cost of internal node
        self.step_peak = [0.0] * total  # out+a+b elems of internal node
        slot = {i: i for i in range(n)}
        nxt = n
        for i, j in toplevel:
            a, b = slot[i], slot[j]
            self._set_children(nxt, a, b)
            slot[i] = nxt
            del slot[j]
            nxt += 1
        assert nxt == total, "toplevel path must contract to one tensor"
        self.root = total - 1

    def _set_children(self, x, a, b):
        self.left[x], self.right[x] = a, b
        self.parent[a] = self.parent[b] = x
        va, vb = self.view[a], self.view[b]
        self.view[x] = va.xor(vb)
        self.op[x] = va.union_size(vb)
        self.step_peak[x] = self.view[x].size + va.size + vb.size

    def total_op(self) -> float:
        return sum(self.op[self.n:])

    def peak(self) -> float:
        return max(self.step_peak[self.n:]) if self.n > 1 else 0.0

    def nni(self, x, child_sel, grand_sel, apply=False):
        """NNI at internal node x: child P = (C, D) (child_sel picks which
        side of x is P), swap x's other child B with C or D (grand_sel).
        Returns (delta_op, new_P_peak) without applying when apply=False."""
        p = self.left[x] if child_sel == 0 else self.right[x]
        b = self.right[x] if child_sel == 0 else self.left[x]
        if p < self.n:
            return None  # P must be internal
        c = self.left[p] if grand_sel == 0 else self.right[p]
        d = self.right[p] if grand_sel == 0 else self.left[p]
        vc, vd, vb = self.view[c], self.view[d], self.view[b]
        # the move changes BOTH intermediates: P' = (C, B) replaces P = (C, D)
        # and X' = (P', D) replaces X = (P, B); view[x] itself is unchanged
        new_p_op = vc.union_size(vb)
        new_p_view = vc.xor(vb)
        new_x_op = new_p_view.union_size(vd)
        delta = (new_p_op + new_x_op) - (self.op[p] + self.op[x])
        new_p_peak = new_p_view.size + vc.size + vb.size
        new_x_peak = self.view[x].size + new_p_view.size + vd.size
        new_peak = max(new_p_peak, new_x_peak)
        if not apply:
            return delta, new_peak
        self.left[p], self.right[p] = c, b
        self.parent[c] = self.parent[b] = p
        self.view[p] = new_p_view
        self.op[p] = new_p_op
        self.step_peak[p] = new_p_peak
        if child_sel == 0:
            self.left[x], self.right[x] = p, d
        else:
            self.left[x], self.right[x] = d, p
        self.parent[p] = self.parent[d] = x
        self.op[x] = new_x_op
        self.step_peak[x] = new_x_peak
        return delta, new_peak

    def to_replace_toplevel(self):
        """Post-order emission to a replace-left pair list over the ORIGINAL
        leaf slots (each internal node's result lives in its leftmost leaf's
        slot, matching ssa_replace_ordering's replace-left convention)."""
        pairs = []

        def walk(node):
            if node < self.n:
                return node
            a = walk(self.left[node])
            b = walk(self.right[node])
            pairs.append((a, b))
            return a

        walk(self.root)
        return pairs


def refine_replace_path(leaves: List[LeafTensor], toplevel, moves=100_000,
                        seed=0, size_cap=None, initial_temperature=None,
                        final_temperature=None):
    """Refine a flat replace-left path by NNI simulated annealing.

    Returns (new_toplevel, op_cost, peak_elems). Never returns a path worse
    than the input (the best-seen tree is kept, reference SA's best-tracking
    semantics, simulated_annealing.rs:137-141)."""
    tree = _Tree(leaves, toplevel)
    rng = np.random.default_rng(seed)
    cur_op = tree.total_op()
    best = (cur_op, tree.peak(), list(toplevel))
    if size_cap is not None and best[1] > size_cap:
        size_cap = best[1]  # never tighten below the input's own peak
    # temperature in units of log2(op ratio), like the reference's log-ratio
    # acceptance; default sweep spans "accept 2x regressions" -> "accept
    # only ~1.02x"
    t0 = 1.0 if initial_temperature is None else initial_temperature
    t1 = 0.03 if final_temperature is None else final_temperature
    n = tree.n
    if n < 3:
        return list(toplevel), cur_op, tree.peak()
    internal = list(range(n, 2 * n - 1))
    log_t0, log_t1 = math.log(t0), math.log(t1)
    for m in range(moves):
        temperature = math.exp(log_t0 + (log_t1 - log_t0) * (m / moves))
        x = internal[int(rng.integers(0, len(internal)))]
        child_sel = int(rng.integers(0, 2))
        grand_sel = int(rng.integers(0, 2))
        res = tree.nni(x, child_sel, grand_sel, apply=False)
        if res is None:
            continue
        delta, new_peak = res
        if size_cap is not None and new_peak > size_cap:
            continue
        new_op = cur_op + delta
        if new_op <= cur_op:
            accept = True
        else:
            # log2-ratio acceptance (simulated_annealing.rs:120-131)
            diff = math.log2(new_op / cur_op)
            accept = math.exp(min(50.0, -diff / temperature)) >= rng.random()
        if not accept:
            continue
        tree.nni(x, child_sel, grand_sel, apply=True)
        cur_op = new_op
        if cur_op < best[0]:
            peak = tree.peak()
            if size_cap is None or peak <= size_cap:
                best = (cur_op, peak, tree.to_replace_toplevel())
    return best[2], best[0], best[1]


def reduce_peak(leaves: List[LeafTensor], toplevel, target_peak,
                moves=100_000, seed=0, initial_temperature=1.0,
                final_temperature=0.02):
    """NNI anneal whose PRIMARY objective is the walk's peak step size
    (out+a+b elements): memory-feasibility rescue for paths whose peak
    exceeds device (or cap) limits. Secondary objective: op count.

    Returns (toplevel, op, peak) of the best (peak, op)-lexicographic tree
    seen. Stops early once peak <= target_peak and a further 25% of the
    move budget brings no improvement."""
    tree = _Tree(leaves, toplevel)
    rng = np.random.default_rng(seed)
    n = tree.n
    if n < 3:
        return list(toplevel), tree.total_op(), tree.peak()
    internal = list(range(n, 2 * n - 1))
    cur_op = tree.total_op()
    cur_peak = tree.peak()
    best = (cur_peak, cur_op, list(toplevel))
    log_t0, log_t1 = math.log(initial_temperature), math.log(final_temperature)
    stale = 0
    for m in range(moves):
        temperature = math.exp(log_t0 + (log_t1 - log_t0) * (m / moves))
        x = internal[int(rng.integers(0, len(internal)))]
        cs = int(rng.integers(0, 2))
        gs = int(rng.integers(0, 2))
        res = tree.nni(x, cs, gs, apply=False)
        if res is None:
            continue
        delta, new_local_peak = res
        if new_local_peak > cur_peak:
            # peak-raising move: SA on the peak ratio
            diff = math.log2(new_local_peak / cur_peak)
            accept = math.exp(min(50.0, -diff / temperature)) >= rng.random()
        elif delta <= 0:
            accept = True
        else:
            diff = math.log2((cur_op + delta) / cur_op)
            accept = math.exp(min(50.0, -diff / temperature)) >= rng.random()
        if not accept:
            continue
        p = tree.left[x] if cs == 0 else tree.right[x]
        old_p_peak, old_x_peak = tree.step_peak[p], tree.step_peak[x]
        tree.nni(x, cs, gs, apply=True)
        cur_op += delta
        if new_local_peak >= cur_peak:
            cur_peak = new_local_peak
        elif max(old_p_peak, old_x_peak) >= cur_peak:
            cur_peak = tree.peak()  # the argmax may have moved: recompute
        stale += 1
        if (cur_peak, cur_op) < (best[0], best[1]):
            best = (cur_peak, cur_op, tree.to_replace_toplevel())
            stale = 0
        if best[0] <= target_peak and stale > moves // 4:
            break
    return best[2], best[1], best[0]


class TreeSA:
    """Pathfinder: PartitionSearch (or a given base finder) followed by NNI
    tree annealing. find_path mirrors the Pathfinder trait (paths.rs:21-28);
    flat networks only (composites are flattened by the caller's tier)."""

    def __init__(self, base=None, moves=100_000, seed=0, size_cap=None,
                 restarts=1):
        self.base = base or PartitionSearch(size_cap=size_cap)
        self.moves = moves
        self.seed = seed
        self.size_cap = size_cap
        self.restarts = restarts

    def find_path(self, tensor: CompositeTensor) -> BasicContractionPathResult:
        from .cost import contract_path_cost

        base = self.base.find_path(tensor)
        flat_ok = all(not isinstance(t, CompositeTensor)
                      for t in tensor.tensors)
        if not flat_ok:
            return base
        leaves = [LeafTensor(t.legs, t.bond_dims) for t in tensor.tensors]
        toplevel = list(base.replace_path().toplevel)
        best_top, best_op, best_peak = toplevel, base.flops, base.size
        for r in range(self.restarts):
            top, op, peak = refine_replace_path(
                leaves, best_top, moves=self.moves, seed=self.seed + r,
                size_cap=self.size_cap)
            if op < best_op:
                best_top, best_op, best_peak = top, op, peak
        replace = ContractionPath.simple(best_top)
        op_cost, mem_cost = contract_path_cost(tensor.tensors, replace, True)
        return BasicContractionPathResult(
            _replace_to_ssa(best_top, len(tensor.tensors)), op_cost, mem_cost)
