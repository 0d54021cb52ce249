"""End-to-end circuit -> network -> (oracle) contraction, pinned against the
reference's exact-value integration tests."""

import math

import numpy as np

from oracle import contract_network
from oracle.adapters import network_to_otensors
from tnc_amd import (
    Circuit,
    CompositeTensor,
    Greedy,
    RandomGreedy,
    TensorData,
    find_partitioning,
    partition_tensor_network,
    random_circuit,
    sycamore_circuit,
)
from tnc_amd.connectivity import ConnectivityLayout, connectivity_edges
from tnc_amd.contraction_path import flatten_network

S2 = 1 / math.sqrt(2)


def oracle_contract(tn, pathfinder=None):
    pathfinder = pathfinder or Greedy()
    result = pathfinder.find_path(tn)
    replace = result.replace_path()
    return contract_network(network_to_otensors(tn), replace)


def test_ghz_statevector():
    """Config 1 / README.md:27-75: GHZ = (1/sqrt2)(|000> + |111>)."""
    c = Circuit()
    qr = c.allocate_register(3)
    c.append_gate(TensorData.from_gate("h"), [qr.qubit(0)])
    c.append_gate(TensorData.from_gate("cx"), [qr.qubit(0), qr.qubit(1)])
    c.append_gate(TensorData.from_gate("cx"), [qr.qubit(1), qr.qubit(2)])
    tn, permutor = c.into_statevector_network()
    out = oracle_contract(tn)
    legs, dims, data = permutor.apply(out.legs, list(out.data.shape), out.data)
    sv = data.reshape(-1)
    ref = np.zeros(8, dtype=np.complex128)
    ref[0] = S2
    ref[7] = S2
    np.testing.assert_allclose(sv, ref, atol=1e-15)


def test_hadamards_amplitude():
    """circuit_builder.rs:362-385: <0|H^5|0> = 2^-2.5."""
    c = Circuit()
    qr = c.allocate_register(5)
    for q in qr.qubits():
        c.append_gate(TensorData.from_gate("h"), [q])
    tn, permutor = c.into_amplitude_network("00000")
    assert permutor.is_identity()
    out = oracle_contract(tn)
    assert out.legs == []
    np.testing.assert_allclose(out.data, S2**5, atol=1e-15)


def test_rx_expectation():
    """circuit_builder.rs:388-415: <Z x Z> = (1/sqrt2) * 0.5."""
    c = Circuit()
    qr = c.allocate_register(2)
    c.append_gate(TensorData.from_gate("rx", [math.pi / 4]), [qr.qubit(0)])
    c.append_gate(TensorData.from_gate("rx", [math.pi / 3]), [qr.qubit(1)])
    tn = c.into_expectation_value_network()
    out = oracle_contract(tn)
    np.testing.assert_allclose(out.data, S2 * 0.5, atol=1e-15)


def _u2(phi, lam):
    # qelib1.inc: u2(phi, lambda) = u(pi/2, phi, lambda)
    return TensorData.from_gate("u", [math.pi / 2, phi, lam])


def test_dj_4qubit_statevector():
    """integration_tests.rs:169-217 (circuit built directly; QASM import is a
    next-row item): result = (1/sqrt2)(|1110> - |1111>)."""
    c = Circuit()
    q = c.allocate_register(4)
    c.append_gate(_u2(0, 0), [q.qubit(0)])
    c.append_gate(_u2(0, 0), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(2)])
    c.append_gate(_u2(-math.pi, -math.pi), [q.qubit(3)])
    c.append_gate(TensorData.from_gate("cx"), [q.qubit(0), q.qubit(3)])
    c.append_gate(_u2(-math.pi, -math.pi), [q.qubit(0)])
    c.append_gate(TensorData.from_gate("cx"), [q.qubit(1), q.qubit(3)])
    c.append_gate(_u2(-math.pi, -math.pi), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("cx"), [q.qubit(2), q.qubit(3)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(2)])
    tn, permutor = c.into_statevector_network()
    out = oracle_contract(tn)
    legs, dims, data = permutor.apply(out.legs, list(out.data.shape), out.data)
    sv = data.reshape(-1)
    ref = np.zeros(16, dtype=np.complex128)
    ref[14] = S2
    ref[15] = -S2
    np.testing.assert_allclose(sv, ref, atol=1e-15)


def test_qft_2qubit_expectation():
    """integration_tests.rs:219-244: expectation = 0.5."""
    c = Circuit()
    q = c.allocate_register(2)
    c.append_gate(TensorData.from_gate("h"), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("cx"), [q.qubit(1), q.qubit(0)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("cp", [math.pi / 2]), [q.qubit(1), q.qubit(0)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(0)])
    c.append_gate(TensorData.from_gate("swap"), [q.qubit(0), q.qubit(1)])
    tn = c.into_expectation_value_network()
    out = oracle_contract(tn, RandomGreedy(3))
    np.testing.assert_allclose(out.data, 0.5, atol=1e-15)


def test_partitioned_equals_unpartitioned():
    """integration_tests.rs:26-86 pattern: partitioning never changes the
    result (15q RQC, depth 10, Eagle)."""
    tn = random_circuit(15, 10, 0.5, 0.5, 52, ConnectivityLayout.EAGLE)
    ref = oracle_contract(tn)

    tn2 = random_circuit(15, 10, 0.5, 0.5, 52, ConnectivityLayout.EAGLE)
    partitioning = find_partitioning(tn2, 4)
    ptn = partition_tensor_network(tn2, partitioning)
    out = oracle_contract(ptn, RandomGreedy(10))
    assert sorted(out.legs) == sorted(ref.legs)
    assert out.legs == ref.legs  # scalar amplitude: both empty
    np.testing.assert_allclose(out.data, ref.data, rtol=1e-10, atol=1e-12)


def test_flatten_matches_recursive():
    """Flat plan == recursive walk on a partitioned network."""
    tn = random_circuit(10, 6, 0.5, 0.5, 7, ConnectivityLayout.EAGLE)
    partitioning = find_partitioning(tn, 3)
    ptn = partition_tensor_network(tn, partitioning)
    result = Greedy().find_path(ptn)
    replace = result.replace_path()
    ref = contract_network(network_to_otensors(ptn), replace)

    leaves, steps, final = flatten_network(ptn, replace)
    from oracle.adapters import leaf_to_otensor

    slots = [leaf_to_otensor(l) for l in leaves]
    from oracle import contract_tensors

    for i, j in steps:
        slots[i] = contract_tensors(slots[i], slots[j])
        slots[j] = None
    np.testing.assert_allclose(slots[final].data, ref.data, atol=1e-12)
    assert slots[final].legs == ref.legs


def test_sycamore_rank_counts():
    """sycamore_circuit.rs:81-98."""
    c = sycamore_circuit(3, 3, 42)
    tn, _ = c.into_amplitude_network("000")
    from collections import Counter

    counts = Counter(len(t.legs) for t in tn.tensors)
    assert counts[1] == 6
    assert counts[2] == 12
    assert counts[4] == 1


def test_eagle_connectivity_shape():
    edges = connectivity_edges(ConnectivityLayout.EAGLE)
    nodes = {u for e in edges for u in e}
    # IBM Eagle: 127 qubits
    assert max(nodes) == 126
    assert len(edges) == len(set(tuple(sorted(e)) for e in edges))


def test_random_circuit_with_set_observable_pinned_legs():
    """random_circuit.rs:358-418: with p=1 and observable on qubit 2, the
    leg layout is fully pinned (RNG-independent)."""
    from tnc_amd.builders import random_circuit_with_set_observable

    tn = random_circuit_with_set_observable(
        4, 3, 1.0, 1.0, [2], 0, ConnectivityLayout.Line(4))
    ref = [
        [0, 1],
        [3, 4, 2, 0], [2, 1, 5, 6], [8, 9, 4, 7], [6, 7, 10, 11],
        [12, 3], [5, 13], [14, 8], [10, 15], [16, 9], [11, 17],
        [19, 20, 18, 12], [18, 13, 21, 22], [23, 24, 20, 14],
        [22, 15, 25, 26], [27, 28, 24, 16], [26, 17, 29, 30],
        [31, 19], [21, 32], [33, 23], [25, 34], [35, 27], [29, 36],
        [37, 28], [30, 38],
        [31], [32], [33], [34], [35], [36], [37], [38],
    ]
    assert len(tn.tensors) == 33
    for t, legs in zip(tn.tensors, ref):
        assert t.legs == legs, (t.legs, legs)


def test_random_circuit_with_observable_pinned_legs():
    """random_circuit.rs:287-355: all probabilities 1 -> 40 tensors with the
    reference's exact leg layout."""
    from tnc_amd.builders import random_circuit_with_observable

    tn = random_circuit_with_observable(
        4, 3, 1.0, 1.0, 1.0, 1, ConnectivityLayout.Line(4))
    assert len(tn.tensors) == 40
    assert tn.tensors[4].legs == [8, 9, 0, 2]
    assert tn.tensors[5].legs == [1, 3, 10, 11]
    assert tn.tensors[-1].legs == [47]


def test_observable_network_contracts():
    """The sandwich network contracts to a scalar, path-independently (the
    reference pairs each side with the SAME unconjugated random state,
    random_circuit.rs:262-270, so the value is complex in general)."""
    from tnc_amd.builders import random_circuit_with_set_observable

    tn = random_circuit_with_set_observable(
        4, 3, 0.7, 0.7, [1, 2], 3, ConnectivityLayout.Line(4))
    out = oracle_contract(tn, RandomGreedy(4))
    assert out.legs == []
    out2 = oracle_contract(tn, Greedy())
    np.testing.assert_allclose(out.data, out2.data, rtol=1e-12)


def test_is_connected():
    from tnc_amd import CompositeTensor, LeafTensor

    bd = {0: 17, 1: 19, 2: 8, 3: 2}
    tn = CompositeTensor([LeafTensor.new_from_map([0, 1], bd),
                          LeafTensor.new_from_map([1, 2], bd)])
    assert tn.is_connected()
    tn.push_tensor(LeafTensor.new_from_map([3], bd))
    assert not tn.is_connected()
