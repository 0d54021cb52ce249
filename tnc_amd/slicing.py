"""Edge slicing: contract a network as a sum over fixed values of chosen
shared edges.

The reference lists slicing as future work (book/src/future_work.md item 2);
none of the benchmark configs need it on a 288 GB MI355X, but it is the
standard memory-for-flops trade once an intermediate outgrows HBM, and a
second parallelism axis beyond partitions (slices are embarrassingly
parallel — each can run on its own GPU with a plain sum at the end).

Fixing one index value of an edge removes that leg from every tensor
carrying it; the SAME replace-left path stays valid (leg sets only shrink,
the symmetric-difference output order of the surviving legs is unchanged),
every intermediate that contained the leg shrinks by its dimension, and the
full result is the elementwise sum of the sliced results over all
assignments. Total flops grow by roughly the product of the sliced dims.
"""

from __future__ import annotations

import itertools
import math

import numpy as np

from .tensor import CompositeTensor, LeafTensor, TensorData


def _leg_dims(tn: CompositeTensor) -> dict:
    dims = {}
    for t in tn.tensors:
        for l, d in zip(t.legs, t.bond_dims):
            dims[l] = d
    return dims


def _walk_sizes(tn: CompositeTensor, replace_toplevel, skip=()):
    """Simulate the replace-left walk (legs only) with `skip` legs removed.
    Returns (peak_elems, list of (size, legs) per intermediate, final legs).
    """
    skip = set(skip)
    views = [
        LeafTensor([l for l in t.legs if l not in skip],
                   [d for l, d in zip(t.legs, t.bond_dims) if l not in skip])
        for t in tn.tensors
    ]
    peak = 0.0
    inters = []
    last = None
    for i, j in replace_toplevel:
        out = views[i] ^ views[j]
        inters.append((out.size(), list(out.legs)))
        peak = max(peak, out.size())
        views[i] = out
        views[j] = None
        last = i
    final_legs = list(views[last].legs) if last is not None else []
    return peak, inters, final_legs


def find_slice_edges(tn: CompositeTensor, replace_toplevel,
                     target_peak_elems: float, max_edges: int = 16):
    """Greedy slice-edge selection: repeatedly slice the shared edge with
    the largest total presence in near-peak intermediates until the
    projected peak intermediate size is <= target_peak_elems. Open legs of
    the final tensor are never sliced (that would change the output).
    Returns (edges, projected_peak_elems)."""
    dims = _leg_dims(tn)
    _, _, final_legs = _walk_sizes(tn, replace_toplevel)
    protected = set(final_legs)
    edges = []
    peak, inters, _ = _walk_sizes(tn, replace_toplevel)
    while len(edges) < max_edges and peak > target_peak_elems:
        score = {}
        for size, legs in inters:
            if size < peak / 4:
                continue
            for l in legs:
                if l in protected or l in edges:
                    continue
                score[l] = score.get(l, 0.0) + size * (1.0 - 1.0 / dims[l])
        if not score:
            break  # nothing left to slice
        edges.append(max(score, key=score.get))
        new_peak, inters, _ = _walk_sizes(tn, replace_toplevel, skip=edges)
        # chasing a memory target: stop when an edge makes no progress.
        # target <= 1 means "give me max_edges edges for parallelism" —
        # an edge in near-peak intermediates still cuts per-slice flops
        # even when another same-size intermediate pins the peak.
        if new_peak >= peak and target_peak_elems > 1:
            edges.pop()
            break
        peak = new_peak
    return edges, peak


def slice_network(tn: CompositeTensor, assignment: dict) -> CompositeTensor:
    """The network with every leg in `assignment` fixed at its index value:
    leaves carrying a sliced leg are materialized and indexed (the leg
    disappears); leaf order is preserved so any path for `tn` stays valid."""
    out = []
    for t in tn.tensors:
        assert isinstance(t, LeafTensor), "slice_network expects a flat network"
        hit = [l for l in t.legs if l in assignment]
        if not hit:
            out.append(t)
            continue
        data = np.asarray(t.tensordata.into_data(), dtype=np.complex128)
        data = data.reshape(tuple(t.bond_dims))
        index = tuple(
            assignment[l] if l in assignment else slice(None) for l in t.legs
        )
        data = data[index]
        legs = [l for l in t.legs if l not in assignment]
        bond = [d for l, d in zip(t.legs, t.bond_dims) if l not in assignment]
        nt = LeafTensor(legs, bond)
        nt.set_tensor_data(TensorData(TensorData.MATRIX, matrix=data))
        out.append(nt)
    return CompositeTensor(out)


def iter_assignments(tn: CompositeTensor, edges):
    dims = _leg_dims(tn)
    for combo in itertools.product(*[range(dims[e]) for e in edges]):
        yield dict(zip(edges, combo))


def num_slices(tn: CompositeTensor, edges) -> int:
    dims = _leg_dims(tn)
    return int(math.prod(dims[e] for e in edges)) if edges else 1


def contract_sliced_gpu(tn: CompositeTensor, replace_path, edges,
                        device: int = 0):
    """Contract `tn` on the GPU as the sum over all slice assignments of
    `edges`. Returns (legs, ndarray) like contract_tensor_network_gpu."""
    from .executor import contract_tensor_network_gpu

    total = None
    legs_out = None
    for assignment in iter_assignments(tn, edges):
        stn = slice_network(tn, assignment)
        legs, data = contract_tensor_network_gpu(stn, replace_path, device)
        if total is None:
            legs_out, total = legs, np.array(data, copy=True)
        else:
            assert legs == legs_out
            total += data
    return legs_out, total
