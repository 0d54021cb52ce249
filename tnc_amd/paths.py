"""Contraction pathfinders, mirroring tnc/src/contractionpath/paths.rs and the
cotengrust Greedy / RandomGreedy methods (contractionpath/paths/cotengrust.rs).

The greedy algorithm is a reimplementation of cotengra's "memory-removed"
greedy (the cotengrust crate, rev 2998e98, is not vendored in the reference):
repeatedly contract the candidate pair with the lowest score
size(ab) - costmod*(size(a)+size(b)), candidates being pairs that share a
leg, then combine remaining disconnected terms by pairwise outer products
smallest-first. Pinned by the reference's expected-path tests
(cotengrust.rs:229-307), reproduced in tests/test_paths.py.

Like the reference (cotengrust.rs:40-91), contraction semantics during the
search are TNC's: the pair's result legs are the symmetric difference.
"""

from __future__ import annotations

import heapq
import math
from typing import List

import numpy as np

from .contraction_path import ContractionPath, ssa_replace_ordering
from .cost import contract_path_cost
from .tensor import CompositeTensor, LeafTensor


class BasicContractionPathResult:
    """paths.rs:45-76."""

    __slots__ = ("ssa_path", "flops", "size")

    def __init__(self, ssa_path, flops, size):
        self.ssa_path = ssa_path
        self.flops = flops
        self.size = size

    def replace_path(self) -> ContractionPath:
        return ssa_replace_ordering(self.ssa_path)

    def __eq__(self, other):
        return (
            self.ssa_path == other.ssa_path
            and self.flops == other.flops
            and self.size == other.size
        )

    def __repr__(self):
        return (
            f"BasicContractionPathResult(ssa_path={self.ssa_path}, "
            f"flops={self.flops}, size={self.size})"
        )


def _greedy_ssa(leaves: List[LeafTensor], costmod=1.0, temperature=0.0, rng=None):
    """One greedy run over flat leaf views; returns an SSA pair list.

    Candidate score: size(ab)/costmod - (size(a)+size(b))*costmod, optionally
    gumbel-perturbed by `temperature` (cotengra's randomized greedy).
    """
    n = len(leaves)
    if n == 0:
        return []
    legs = {i: t for i, t in enumerate(leaves)}  # ssa id -> LeafTensor view
    sizes = {i: leaves[i].size() for i in range(n)}
    # leg -> set of live ssa ids
    leg_nodes = {}
    for i, t in enumerate(leaves):
        for l in t.legs:
            leg_nodes.setdefault(l, set()).add(i)

    def perturb(score):
        if temperature and rng is not None:
            u = rng.random()
            score = score - temperature * (-math.log(max(-math.log(max(u, 1e-300)), 1e-300)))
        return score

    heap = []
    counter = 0

    def push_candidate(i, j):
        nonlocal counter
        tij = legs[i] ^ legs[j]
        score = tij.size() / costmod - (sizes[i] + sizes[j]) * costmod
        heapq.heappush(heap, (perturb(score), counter, i, j, tij))
        counter += 1

    seen_pairs = set()
    for i in range(n):
        for l in leaves[i].legs:
            for j in leg_nodes[l]:
                if j > i and (i, j) not in seen_pairs:
                    seen_pairs.add((i, j))
                    push_candidate(i, j)

    ssa_path = []
    next_id = n
    alive = set(range(n))
    while heap:
        _, _, i, j, tij = heapq.heappop(heap)
        if i not in alive or j not in alive:
            continue
        # contract i, j -> new node
        new = next_id
        next_id += 1
        ssa_path.append((i, j))
        alive.discard(i)
        alive.discard(j)
        alive.add(new)
        legs[new] = tij
        sizes[new] = tij.size()
        neighbors = set()
        for l in tij.legs:
            s = leg_nodes.setdefault(l, set())
            s.discard(i)
            s.discard(j)
            s.add(new)
            neighbors |= {x for x in s if x != new and x in alive}
        # drop i/j from legs they no longer carry
        for t, old in ((legs.get(i), i), (legs.get(j), j)):
            if t is not None:
                for l in t.legs:
                    leg_nodes.get(l, set()).discard(old)
        for nb in neighbors:
            push_candidate(new, nb)

    # remaining terms are pairwise disjoint: combine by outer products,
    # smallest sizes first; ties pop the LARGER ssa id first (matches the
    # reference's pinned outer-product paths, cotengrust.rs:262-291).
    rest = [(sizes[i], -i, i) for i in alive]
    heapq.heapify(rest)
    while len(rest) > 1:
        _, _, a = heapq.heappop(rest)
        _, _, b = heapq.heappop(rest)
        new = next_id
        next_id += 1
        ssa_path.append((a, b))
        tab = legs[a] ^ legs[b]
        legs[new] = tab
        sizes[new] = tab.size()
        heapq.heappush(rest, (sizes[new], -new, new))
    return ssa_path


class _CotengrustLike:
    """Shared find_path recursion (cotengrust.rs:120-155)."""

    def _optimize_single(self, leaves, external):
        raise NotImplementedError

    def find_path(self, tensor: CompositeTensor) -> BasicContractionPathResult:
        nested_paths = {}
        flat_leaves = []
        for index, t in enumerate(tensor.tensors):
            if isinstance(t, CompositeTensor):
                sub = self.__class__(**self._ctor_args())
                result = sub.find_path(t)
                nested_paths[index] = result.ssa_path
                flat_leaves.append(t.external_tensor())
            else:
                flat_leaves.append(t)
        external = tensor.external_tensor()
        outer = self._optimize_single(flat_leaves, external)
        best = ContractionPath(nested=nested_paths, toplevel=outer)
        replace = ssa_replace_ordering(best)
        op_cost, mem_cost = contract_path_cost(tensor.tensors, replace, True)
        return BasicContractionPathResult(best, op_cost, mem_cost)

    def _ctor_args(self):
        return {}


class Greedy(_CotengrustLike):
    """OptMethod::Greedy (cotengrust.rs:51-61): deterministic greedy."""

    def _optimize_single(self, leaves, external):
        return _greedy_ssa(leaves)


class RandomGreedy(_CotengrustLike):
    """OptMethod::RandomGreedy(n) (cotengrust.rs:62-76): n randomized greedy
    trials (costmod log-uniform in [0.1, 4], temperature log-uniform in
    [0.001, 1]); keeps the path with the lowest op count. Seeded (default 42,
    like cotengrust.rs:71)."""

    def __init__(self, ntrials, seed=42, size_cap=None):
        self.ntrials = ntrials
        self.seed = seed
        # optional peak-memory cap (elements, contract_size_tensors metric):
        # among trials within the cap, the lowest op count wins; with none
        # within the cap, the smallest peak wins. The reference has no such
        # cap (its CPU runs page instead); on a 288 GB GPU it is load-bearing.
        self.size_cap = size_cap

    def _ctor_args(self):
        return {"ntrials": self.ntrials, "seed": self.seed,
                "size_cap": self.size_cap}

    def _optimize_single(self, leaves, external):
        rng = np.random.Generator(np.random.PCG64(self.seed))
        best_path = None
        best_key = None
        # trial 0 is the deterministic greedy; keeps RandomGreedy >= Greedy
        for trial in range(max(1, self.ntrials)):
            if trial == 0:
                costmod, temperature = 1.0, 0.0
            else:
                costmod = math.exp(rng.uniform(math.log(0.1), math.log(4.0)))
                temperature = math.exp(rng.uniform(math.log(0.001), math.log(1.0)))
            ssa = _greedy_ssa(leaves, costmod, temperature, rng)
            cost, peak = _ssa_op_cost(leaves, ssa)
            if self.size_cap is not None:
                key = (peak > self.size_cap, peak if peak > self.size_cap else cost)
            else:
                key = (False, cost)
            if best_key is None or key < best_key:
                best_key = key
                best_path = ssa
        return best_path or []


def _ssa_op_cost(leaves, ssa_path):
    """(op count, peak size) of an SSA path over leaves — the op count is
    prod-of-union-dims per step (contraction_cost.rs:49-52), the peak is
    out+a+b elements (contraction_cost.rs:69-72)."""
    views = list(leaves)
    cost = 0.0
    peak = 0.0
    for i, j in ssa_path:
        ti, tj = views[i], views[j]
        cost += (ti | tj).size()
        out = ti ^ tj
        peak = max(peak, out.size() + ti.size() + tj.size())
        views.append(out)
    return cost, peak
