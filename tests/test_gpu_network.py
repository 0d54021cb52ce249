"""Full-network contraction on the GPU (tn_net executor) vs the oracle and
the reference's exact values."""

import math

import numpy as np
import pytest

from oracle import contract_network
from oracle.adapters import network_to_otensors

pytestmark = pytest.mark.gpu

S2 = 1 / math.sqrt(2)


def gpu_contract(tn, pathfinder=None):
    from tnc_amd import Greedy
    from tnc_amd.executor import contract_tensor_network_gpu

    pathfinder = pathfinder or Greedy()
    replace = pathfinder.find_path(tn).replace_path()
    legs, data = contract_tensor_network_gpu(tn, replace)
    ref = contract_network(network_to_otensors(tn), replace)
    return legs, data, ref


def test_ghz_statevector():
    from tnc_amd import Circuit, TensorData

    c = Circuit()
    qr = c.allocate_register(3)
    c.append_gate(TensorData.from_gate("h"), [qr.qubit(0)])
    c.append_gate(TensorData.from_gate("cx"), [qr.qubit(0), qr.qubit(1)])
    c.append_gate(TensorData.from_gate("cx"), [qr.qubit(1), qr.qubit(2)])
    tn, permutor = c.into_statevector_network()
    legs, data, ref = gpu_contract(tn)
    assert legs == ref.legs
    np.testing.assert_allclose(data, ref.data, atol=1e-15)
    _, _, out = permutor.apply(legs, list(data.shape), data)
    sv = out.reshape(-1)
    expect = np.zeros(8, dtype=np.complex128)
    expect[0] = expect[7] = S2
    np.testing.assert_allclose(sv, expect, atol=1e-15)


def test_hadamards_amplitude():
    from tnc_amd import Circuit, TensorData

    c = Circuit()
    qr = c.allocate_register(5)
    for q in qr.qubits():
        c.append_gate(TensorData.from_gate("h"), [q])
    tn, _ = c.into_amplitude_network("00000")
    legs, data, ref = gpu_contract(tn)
    assert legs == []
    np.testing.assert_allclose(data, S2**5, atol=1e-15)


def test_qft_expectation():
    from tnc_amd import Circuit, RandomGreedy, TensorData

    c = Circuit()
    q = c.allocate_register(2)
    c.append_gate(TensorData.from_gate("h"), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("cx"), [q.qubit(1), q.qubit(0)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(1)])
    c.append_gate(TensorData.from_gate("cp", [math.pi / 2]), [q.qubit(1), q.qubit(0)])
    c.append_gate(TensorData.from_gate("h"), [q.qubit(0)])
    c.append_gate(TensorData.from_gate("swap"), [q.qubit(0), q.qubit(1)])
    tn = c.into_expectation_value_network()
    legs, data, ref = gpu_contract(tn, RandomGreedy(3))
    np.testing.assert_allclose(data, 0.5, atol=1e-15)


def test_rqc24_fixture_vs_oracle():
    """Config 2: the frozen 24q RQC amplitude network, frozen path."""
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.executor import contract_tensor_network_gpu
    from tnc_amd.fixtures import load_fixture

    tn, replace_toplevel, meta = load_fixture("rqc24")
    replace = ContractionPath.simple(replace_toplevel)
    legs, data = contract_tensor_network_gpu(tn, replace)
    ref = contract_network(network_to_otensors(tn), replace)
    assert legs == ref.legs
    # single amplitude: scalar; 1e-10 relative bar from the north star
    np.testing.assert_allclose(data, ref.data, rtol=1e-10)


def test_engine_repeatable():
    """Leaves persist; repeated contract gives identical results."""
    from tnc_amd import Greedy
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.builders import random_circuit
    from tnc_amd.executor import ContractionEngine

    tn = random_circuit(12, 8, 0.5, 0.5, 5, ConnectivityLayout.EAGLE)
    replace = Greedy().find_path(tn).replace_path()
    eng = ContractionEngine(tn, replace)
    try:
        eng.contract()
        _, first = eng.result()
        eng.contract()
        _, second = eng.result()
        np.testing.assert_array_equal(first, second)
        ref = contract_network(network_to_otensors(tn), replace)
        np.testing.assert_allclose(first, ref.data, rtol=1e-10)
    finally:
        eng.close()


def test_engine_profiled():
    from tnc_amd import Greedy
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.builders import random_circuit
    from tnc_amd.executor import ContractionEngine

    tn = random_circuit(10, 6, 0.5, 0.5, 9, ConnectivityLayout.EAGLE)
    replace = Greedy().find_path(tn).replace_path()
    eng = ContractionEngine(tn, replace)
    try:
        elapsed, step_ms, gemm_ms, kinds = eng.contract_profiled()
        assert len(step_ms) == len(eng.steps)
        assert all(m >= 0.0 for m in step_ms)
        assert elapsed > 0
    finally:
        eng.close()


def test_fails_loudly_without_extension(monkeypatch):
    """The product path must never fall back: a missing .so raises."""
    import importlib

    import tnc_amd.hiplib as hiplib

    monkeypatch.setattr(hiplib, "_LIB_PATH", "/nonexistent/libtnc_hip.so")
    monkeypatch.setattr(hiplib, "_lib", None)
    with pytest.raises(RuntimeError, match="not built"):
        hiplib.lib()


def test_syc49_engine_builds():
    """Config 5 fixture loads and a prefix of the frozen path runs in c64
    (the full contraction is the bench's job; here: plumbing + parity on
    the first 400 steps against the oracle)."""
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture

    tn, rp, meta = load_fixture("syc49")
    assert meta.get("dtype") == "c64"
    prefix = rp[:400]
    # a prefix is not a full contraction; engine asserts full contraction,
    # so test via per-step einsum comparison instead on a sub-walk
    from oracle.adapters import network_to_otensors
    from oracle import contract_tensors
    import tnc_amd.hiplib as hiplib

    slots = network_to_otensors(tn)
    for s, (i, j) in enumerate(prefix[:40]):
        ref = contract_tensors(slots[i], slots[j])
        got = hiplib.einsum_c64(ref.legs, slots[i].legs,
                                slots[i].data.astype(np.complex64),
                                slots[j].legs,
                                slots[j].data.astype(np.complex64))
        np.testing.assert_allclose(got, ref.data.astype(np.complex64),
                                   rtol=2e-3, atol=1e-4)
        slots[i] = ref
        slots[j] = None


def test_graph_replay_identical():
    """contract() three times: normal run, hipGraph capture+replay, pure
    replay — every pass must match the oracle (and each other bit-exactly).
    Uses the rqc24 fixture so the engine reserves an arena (graph capture
    only arms with one)."""
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture

    from tnc_amd import hiplib

    tn, rp, meta = load_fixture("rqc24")
    replace = ContractionPath.simple(rp)
    ref = contract_network(network_to_otensors(tn), replace)
    eng = ContractionEngine(tn, replace)
    # rqc24 is below the executor's auto-reserve threshold; force an arena
    # so graph capture arms (it needs stable workspace addresses)
    hiplib.check(hiplib.lib().tn_net_reserve(eng.net, 256 * 1024 * 1024),
                 "tn_net_reserve")
    try:
        results = []
        for _ in range(3):
            eng.contract()
            _, data = eng.result()
            results.append(np.array(data, copy=True))
        for r in results:
            np.testing.assert_allclose(r, ref.data, rtol=1e-10, atol=1e-12)
        assert (results[0] == results[1]).all()
        assert (results[1] == results[2]).all()
    finally:
        eng.close()


def test_qasm_derived_network_gpu():
    """North-star parity sentence made literal: a QASM-derived amplitude
    network contracts on the GPU to the oracle's value within 1e-10
    relative (c128)."""
    from tnc_amd import RandomGreedy
    from tnc_amd.qasm import import_qasm

    lines = ['OPENQASM 2.0;', 'include "qelib1.inc";', "qreg q[10];"]
    for i in range(10):
        lines.append(f"h q[{i}];")
    for i in range(9):
        lines.append(f"cx q[{i}],q[{i+1}];")
    for i in range(10):
        lines.append(f"u3(0.3,{0.1*i},-0.2) q[{i}];")
    for i in range(0, 9, 2):
        lines.append(f"cp(pi/{i+2}) q[{i}],q[{i+1}];")
    for i in range(10):
        lines.append(f"t q[{i}];")
    lines.append("swap q[0],q[9];")
    circuit = import_qasm("\n".join(lines))
    tn, _ = circuit.into_amplitude_network("0" * 10)
    legs, data, ref = gpu_contract(tn, RandomGreedy(8))
    np.testing.assert_allclose(data, ref.data, rtol=1e-10)


def test_sliced_contraction_gpu():
    """Slicing (memory-for-flops): sum of GPU-contracted slices equals the
    direct contraction and the oracle."""
    from tnc_amd import Greedy
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.executor import contract_tensor_network_gpu
    from tnc_amd.slicing import contract_sliced_gpu, find_slice_edges, \
        _walk_sizes

    tn = random_circuit(16, 14, 0.5, 0.8, 3, ConnectivityLayout.EAGLE)
    replace = Greedy().find_path(tn).replace_path()
    peak, _, _ = _walk_sizes(tn, replace.toplevel)
    edges, new_peak = find_slice_edges(tn, replace.toplevel, peak / 4,
                                       max_edges=4)
    assert edges and new_peak < peak
    legs_s, sliced = contract_sliced_gpu(tn, replace, edges)
    legs_d, direct = contract_tensor_network_gpu(tn, replace)
    ref = contract_network(network_to_otensors(tn), replace)
    assert legs_s == legs_d == ref.legs
    np.testing.assert_allclose(sliced, direct, rtol=1e-12, atol=1e-13)
    np.testing.assert_allclose(sliced, ref.data, rtol=1e-10, atol=1e-12)


def test_peps_gpu_vs_oracle():
    """Data-filled PEPS sandwich (2x3, one PEPO layer) contracts on the GPU
    to the oracle's scalar."""
    from tnc_amd import Greedy
    from tnc_amd.builders import peps
    from tnc_amd.executor import contract_tensor_network_gpu
    from tnc_amd.tensor import TensorData

    rng = np.random.default_rng(5)
    tn = peps(2, 3, 2, 2, 1)
    for t in tn.tensors:
        data = (rng.standard_normal(t.shape) +
                1j * rng.standard_normal(t.shape))
        t.set_tensor_data(TensorData(TensorData.MATRIX, matrix=data))
    replace = Greedy().find_path(tn).replace_path()
    legs, data = contract_tensor_network_gpu(tn, replace)
    ref = contract_network(network_to_otensors(tn), replace)
    assert legs == [] == ref.legs
    np.testing.assert_allclose(data, ref.data, rtol=1e-10)


def test_engine_fuzz_random_circuits():
    """Engine-level fuzz: random circuits of varying density/seed through
    the full device walk (prepack overlap, TTGT routing, graph arming)
    against the oracle. Broader coverage for the cross-step machinery
    than the fixed fixtures."""
    from tnc_amd import RandomGreedy
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.executor import ContractionEngine

    cases = [
        (13, 9, 0.4, 0.7, 101),
        (14, 7, 0.6, 0.5, 102),
        (12, 11, 0.5, 0.8, 103),
        (15, 8, 0.3, 0.6, 104),
        (11, 12, 0.7, 0.7, 105),
        (16, 6, 0.5, 0.4, 106),
    ]
    for q, r, p1, p2, seed in cases:
        tn = random_circuit(q, r, p1, p2, seed, ConnectivityLayout.EAGLE)
        replace = RandomGreedy(4, seed=seed).find_path(tn).replace_path()
        ref = contract_network(network_to_otensors(tn), replace)
        eng = ContractionEngine(tn, replace)
        try:
            eng.contract()
            _, first = eng.result()
            eng.contract()  # second pass: graph capture+replay path
            _, second = eng.result()
        finally:
            eng.close()
        np.testing.assert_allclose(first, ref.data, rtol=1e-10, atol=1e-12,
                                   err_msg=f"case {(q, r, p1, p2, seed)}")
        np.testing.assert_array_equal(first, second)
