"""Edge slicing (memory-for-flops; reference future_work item 2): the sum
of sliced contractions equals the direct contraction, the same path serves
every slice, and the greedy selector actually reduces the projected peak."""

import numpy as np

from oracle import contract_network
from oracle.adapters import network_to_otensors
from tnc_amd import Greedy
from tnc_amd.builders import random_circuit
from tnc_amd.connectivity import ConnectivityLayout
from tnc_amd.slicing import (find_slice_edges, iter_assignments, num_slices,
                             slice_network, _walk_sizes)


def _net_and_path(qubits=16, rounds=14, seed=3, p2=0.8):
    tn = random_circuit(qubits, rounds, 0.5, p2, seed, ConnectivityLayout.EAGLE)
    result = Greedy().find_path(tn)
    return tn, result.replace_path()


def test_slice_sum_equals_direct():
    tn, replace = _net_and_path()
    direct = contract_network(network_to_otensors(tn), replace)

    peak, _, _ = _walk_sizes(tn, replace.toplevel)
    edges, new_peak = find_slice_edges(tn, replace.toplevel, peak / 4,
                                       max_edges=4)
    assert edges, "expected at least one sliced edge"
    assert new_peak < peak

    total = None
    for assignment in iter_assignments(tn, edges):
        stn = slice_network(tn, assignment)
        part = contract_network(network_to_otensors(stn), replace)
        if total is None:
            legs, total = part.legs, np.array(part.data, copy=True)
        else:
            assert part.legs == legs
            total += part.data
    assert legs == direct.legs
    np.testing.assert_allclose(total, direct.data, rtol=1e-12, atol=1e-14)


def test_num_slices_and_assignments():
    tn, replace = _net_and_path(qubits=16, rounds=14, seed=3)
    peak, _, _ = _walk_sizes(tn, replace.toplevel)
    edges, _ = find_slice_edges(tn, replace.toplevel, peak / 8, max_edges=4)
    n = num_slices(tn, edges)
    assert n == 2 ** len(edges)  # all-qubit networks: dims are 2
    assert sum(1 for _ in iter_assignments(tn, edges)) == n


def test_final_legs_never_sliced():
    tn = random_circuit(8, 6, 0.5, 0.5, 11, ConnectivityLayout.EAGLE)
    # statevector-style: leave open legs by not closing the circuit — the
    # builder network is amplitude-closed, so instead protect via a tiny
    # target that wants to slice everything and check the final legs survive
    replace = Greedy().find_path(tn).replace_path()
    _, _, final_legs = _walk_sizes(tn, replace.toplevel)
    edges, _ = find_slice_edges(tn, replace.toplevel, 1.0, max_edges=8)
    assert not set(edges) & set(final_legs)


def test_slice_network_keeps_order():
    tn, replace = _net_and_path()
    edges, _ = find_slice_edges(
        tn, replace.toplevel,
        _walk_sizes(tn, replace.toplevel)[0] / 2, max_edges=2)
    if not edges:
        return
    stn = slice_network(tn, {edges[0]: 1})
    assert len(stn.tensors) == len(tn.tensors)
    for old, new in zip(tn.tensors, stn.tensors):
        assert [l for l in old.legs if l != edges[0]] == list(new.legs)
