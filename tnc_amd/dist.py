"""Partition-parallel distributed contraction: one partition per GPU,
RCCL send/recv of open-leg intermediates over xGMI.

Mirrors tnc/src/mpi/communication.rs:
- rank<->partition mapping: the partition holding the final tensor goes to
  rank 0, the rest in order (get_tensor_mapping, communication.rs:89-115);
- local phase: each rank contracts its partition independently (zero
  communication);
- fan-in (intermediate_reduce_tensor_network, communication.rs:199-249): for
  each toplevel pair (x, y), rank(y) sends its contracted leaf to rank(x),
  which contracts {local, received} as a single pair; the final tensor is
  forwarded to rank 0 (communication.rs:236-247).

Unlike the reference there is NO serialization: every rank derives the full
plan (partitioning, paths, all intermediate leg lists and shapes)
deterministically from the shared fixture, so the wire carries only raw c128
device buffers. Tensors travel as float64 views (re, im interleaved), which
both the gloo (CPU tests) and nccl/RCCL backends support.

The contraction backends are injected so the orchestration is testable on
CPU: the GPU backend lives in bench.py / executor.py; tests inject an
oracle-based one.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Tuple

from .contraction_path import ContractionPath
from .cost import contract_cost_tensors
from .partition import find_partitioning, partition_tensor_network
from .paths import RandomGreedy
from .tensor import CompositeTensor, LeafTensor


def _local_final_view(part, inner: "ContractionPath | None") -> LeafTensor:
    """Leg list/order of a partition's local contraction result: simulate
    the replace-left walk over metadata (contraction.rs:52-57 semantics)."""
    if not isinstance(part, CompositeTensor):
        return LeafTensor(part.legs, part.bond_dims)
    views = [
        _local_final_view(t, inner.nested.get(i) if inner else None)
        if isinstance(t, CompositeTensor)
        else LeafTensor(t.legs, t.bond_dims)
        for i, t in enumerate(part.tensors)
    ]
    if inner is None or not inner.toplevel:
        assert len(views) == 1
        return views[0]
    for i, j in inner.toplevel:
        views[i] = views[i] ^ views[j]
        views[j] = None
    remaining = [v for v in views if v is not None]
    assert len(remaining) == 1
    return remaining[0]


class DistPlan:
    """Deterministic distributed plan shared by all ranks."""

    def __init__(self, partitioned: CompositeTensor, path: ContractionPath,
                 nranks: int):
        assert len(partitioned.tensors) <= nranks or True
        self.partitioned = partitioned
        self.path = path
        self.nparts = len(partitioned.tensors)
        # rank mapping (communication.rs:89-115): final partition -> rank 0
        if path.toplevel:
            final_part = path.toplevel[-1][0]
        else:
            final_part = 0
        self.part_rank: Dict[int, int] = {final_part: 0}
        nxt = 1
        for idx in sorted(path.nested.keys()):
            if idx != final_part:
                self.part_rank[idx] = nxt
                nxt += 1
        for idx in range(self.nparts):  # parts without nested path (1 leaf)
            if idx not in self.part_rank:
                self.part_rank[idx] = nxt
                nxt += 1
        assert nxt <= nranks, f"need {nxt} ranks, got {nranks}"
        self.used_ranks = nxt
        # Open-leg view of each partition IN THE ORDER the local replace-left
        # walk actually produces (simulated statically from the nested path)
        # — the wire carries raw buffers, so leg order must be derived, not
        # assumed (the reference ships leg metadata on the wire instead,
        # serialization.rs; here the plan is deterministic on every rank).
        self.externals: List[LeafTensor] = [
            _local_final_view(t, path.nested.get(idx))
            for idx, t in enumerate(partitioned.tensors)
        ]
        # fan-in flops (metric numerator share of the exchange phase)
        views = [LeafTensor(t.legs, t.bond_dims) for t in self.externals]
        self.fanin_flops = 0.0
        self.fanin_shapes: List[Tuple[List[int], List[int]]] = []
        for x, y in path.toplevel:
            self.fanin_flops += contract_cost_tensors(views[x], views[y])
            out = views[x] ^ views[y]
            self.fanin_shapes.append((list(views[y].legs), list(views[y].bond_dims)))
            views[x] = out
        self.final_legs = views[path.toplevel[-1][0]].legs if path.toplevel else (
            self.externals[0].legs)

    def local_flops(self, part: int) -> float:
        sub = self.partitioned.tensors[part]
        if not isinstance(sub, CompositeTensor):
            return 0.0
        inner = self.path.nested.get(part)
        if inner is None:
            return 0.0
        views = [LeafTensor(t.legs, t.bond_dims) for t in sub.tensors]
        total = 0.0
        for i, j in inner.toplevel:
            total += contract_cost_tensors(views[i], views[j])
            views[i] = views[i] ^ views[j]
        return total

    def total_flops(self) -> float:
        return sum(self.local_flops(p) for p in range(self.nparts)) + self.fanin_flops


def make_tree_plan(tn: CompositeTensor, replace_toplevel, nranks: int) -> DistPlan:
    """Partition by cutting the (frozen) contraction tree: repeatedly split
    the heaviest subtree until `nranks` subtrees remain. Each subtree's
    leaves form one partition contracted locally by the subtree's own steps;
    the upper tree becomes the fan-in path. Total executed flops equal the
    single-GPU path's, and every exchanged tensor is an intermediate the
    1-GPU plan already materializes (so memory is bounded by the frozen
    path's own peak) — unlike a fresh k-way min-cut, whose boundary tensors
    on amplitude networks can blow up astronomically.

    The reference reaches feasible plans via KaHyPar + SA instead
    (repartitioning); this construction is an MI355X-side improvement with
    identical results. The serial heaviest step still bounds the critical
    path (the reference has no slicing either; book/src/future_work.md)."""
    n = len(tn.tensors)
    assert nranks >= 2
    # contraction tree from the replace-left path: node = (kind, payload)
    slot_node = {i: ("leaf", i) for i in range(n)}
    views = [LeafTensor(t.legs, t.bond_dims) for t in tn.tensors]
    nodes = []  # internal: dict(left, right, flops_subtree, step_index)
    node_views = {}

    def node_flops(node):
        return 0.0 if node[0] == "leaf" else nodes[node[1]]["flops"]

    def view_of(node):
        return (views[node[1]] if node[0] == "leaf"
                else nodes[node[1]]["view"])

    for step_idx, (i, j) in enumerate(replace_toplevel):
        a, b = slot_node[i], slot_node[j]
        va, vb = view_of(a), view_of(b)
        out = va ^ vb
        nodes.append({
            "left": a, "right": b, "view": out,
            "flops": node_flops(a) + node_flops(b)
                     + contract_cost_tensors(va, vb),
            "step": step_idx,
        })
        slot_node[i] = ("node", len(nodes) - 1)
        slot_node.pop(j)
    (root,) = slot_node.values()

    # split the heaviest subtree until we have nranks components
    components = [root]
    while len(components) < nranks:
        components.sort(key=node_flops, reverse=True)
        heavy = components[0]
        if heavy[0] == "leaf":
            break  # cannot split further
        components = components[1:] + [nodes[heavy[1]]["left"],
                                       nodes[heavy[1]]["right"]]
    # leaves of each component, in global order
    def collect(node, out):
        if node[0] == "leaf":
            out.append(node[1])
        else:
            collect(nodes[node[1]]["left"], out)
            collect(nodes[node[1]]["right"], out)

    comp_leaves = []
    for comp in components:
        ls = []
        collect(comp, ls)
        comp_leaves.append(sorted(ls))
    # partitioning vector; partition ids in component order
    partitioning = [0] * n
    for pid, ls in enumerate(comp_leaves):
        for g in ls:
            partitioning[g] = pid
    ptn = partition_tensor_network(tn, partitioning)
    # partition order in ptn = first appearance of pid; remap to that order
    order = []
    for p in partitioning:
        if p not in order:
            order.append(p)
    pos = {p: i for i, p in enumerate(order)}

    # local (nested) paths: original steps whose node lies inside a
    # component, re-indexed to local leaf positions; fan-in = the rest,
    # in original step order, over partition positions.
    comp_of_node = {}
    for pid, comp in enumerate(components):
        def mark(node, pid=pid):
            if node[0] == "node":
                comp_of_node[node[1]] = pid
                mark(nodes[node[1]]["left"])
                mark(nodes[node[1]]["right"])
        if comp[0] == "node":
            mark(comp)
    local_index = {}  # global leaf -> (pid, local idx)
    for pid, ls in enumerate(comp_leaves):
        for li, g in enumerate(ls):
            local_index[g] = (pid, li)

    nested = {}
    # walk the original path again tracking, per node, its "slot": either
    # (pid, local slot) while inside a component, or partition position at
    # and above the cut
    slot_of = {}
    for i in range(n):
        slot_of[("leaf", i)] = ("local", *local_index[i])
    toplevel = []
    for idx, nd in enumerate(nodes):
        a, b = slot_of[nd["left"]], slot_of[nd["right"]]
        if idx in comp_of_node:
            pid = comp_of_node[idx]
            assert a[0] == "local" and b[0] == "local"
            nested.setdefault(pos[pid], ContractionPath()).toplevel.append(
                (a[2], b[2]))
            slot_of[("node", idx)] = a
        else:
            # fan-in merge: operands are component results
            def part_pos(s):
                if s[0] == "local":
                    return pos[s[1]]
                return s[1]
            pa, pb = part_pos(a), part_pos(b)
            toplevel.append((pa, pb))
            slot_of[("node", idx)] = ("part", pa)
    # single-leaf components appear only via their leaf slots; components
    # that are whole subtrees already produced their final at their root.
    # Components never merged (nranks == 1 case) cannot happen here.
    path = ContractionPath(nested=nested, toplevel=toplevel)
    return DistPlan(ptn, path, nranks)


def make_plan(tn: CompositeTensor, nranks: int, trials: int = 16,
              size_cap=None, seed: int = 0, sa_rounds: int = 0) -> DistPlan:
    """Partition + per-partition paths + fan-in path, all deterministic.

    sa_rounds > 0 refines the initial min-cut partitioning with the
    reference's simulated-annealing repartitioner (simulated_annealing.rs
    semantics; see repartition.py) before path finding, running exactly
    sa_rounds SA rounds so that independent ranks seeding the same rng
    derive the SAME plan (a wall-clock budget would make the plan
    timing-dependent and desynchronize rank wire shapes)."""
    if nranks == 1:
        raise ValueError("use the single-GPU engine for one rank")
    partitioning = find_partitioning(tn, nranks, seed=seed)
    if sa_rounds > 0:
        import numpy as np

        from .repartition import NaivePartitioningModel, balance_partitions

        model = NaivePartitioningModel(tn, nranks)
        partitioning, _ = balance_partitions(
            model, partitioning, np.random.default_rng(seed),
            n_rounds=sa_rounds
        )
    ptn = partition_tensor_network(tn, partitioning)
    result = RandomGreedy(trials, size_cap=size_cap).find_path(ptn)
    return DistPlan(ptn, result.replace_path(), nranks)


def run_fanin(plan: DistPlan, rank: int,
              local: "object",
              send: Callable[[object, List[int], int], None],
              recv: Callable[[List[int], List[int], int], object],
              contract_pair: Callable[[object, List[int], List[int],
                                       object, List[int], List[int]], object]):
    """The fan-in walk (communication.rs:199-249). `local` is this rank's
    contracted partition handle (backend-specific). Returns the final handle
    on rank 0 (None elsewhere).

    send(handle, legs, dims, peer); recv(legs, dims, peer) -> handle;
    contract_pair(a, a_legs, a_dims, b, b_legs, b_dims) -> handle.
    """
    views = [LeafTensor(t.legs, t.bond_dims) for t in plan.externals]
    final_rank = 0
    for x, y in plan.path.toplevel:
        receiver = plan.part_rank[x]
        sender = plan.part_rank[y]
        final_rank = receiver
        if receiver == rank:
            received = recv(views[y].legs, views[y].bond_dims, sender)
            local = contract_pair(local, views[x].legs, views[x].bond_dims,
                                  received, views[y].legs, views[y].bond_dims)
        if sender == rank:
            send(local, views[y].legs, views[y].bond_dims, receiver)
            local = None
        views[x] = views[x] ^ views[y]
    # forward final tensor to rank 0 (communication.rs:236-247)
    if final_rank != 0:
        fx = plan.path.toplevel[-1][0] if plan.path.toplevel else 0
        if rank == final_rank:
            send(local, views[fx].legs, views[fx].bond_dims, 0)
            local = None
        elif rank == 0:
            local = recv(views[fx].legs, views[fx].bond_dims, final_rank)
    return local if rank == 0 else None
