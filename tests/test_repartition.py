"""SA repartitioning (repartitioning.rs / simulated_annealing.rs semantics)."""

import numpy as np

from oracle import contract_network
from oracle.adapters import network_to_otensors
from tnc_amd import (
    NaivePartitioningModel,
    balance_partitions,
    compute_solution,
    find_partitioning,
)
from tnc_amd.builders import random_circuit
from tnc_amd.connectivity import ConnectivityLayout
from tnc_amd.contraction_path import validate_path


def _net():
    return random_circuit(12, 8, 0.5, 0.5, 3, ConnectivityLayout.EAGLE)


def test_compute_solution_structure():
    tn = _net()
    part = find_partitioning(tn, 4)
    ptn, path, parallel, total = compute_solution(tn, part)
    assert parallel <= total + 1e-9
    assert len(path.toplevel) == len(ptn.tensors) - 1
    validate_path(path)
    # the composed plan still contracts to the reference result
    ref = contract_network(
        network_to_otensors(tn),
        __import__("tnc_amd").Greedy().find_path(tn).replace_path(),
    )
    out = contract_network(network_to_otensors(ptn), path)
    np.testing.assert_allclose(out.data, ref.data, rtol=1e-10, atol=1e-12)


def test_sa_improves_or_equal():
    tn = _net()
    k = 4
    initial = find_partitioning(tn, k)
    model = NaivePartitioningModel(tn, k)
    rng = np.random.default_rng(0)
    init_score = model.evaluate(initial, rng)
    best, best_score = balance_partitions(
        model, initial, np.random.default_rng(0), max_time_s=1.5,
        n_trials=2, n_steps=8,
    )
    assert best_score <= init_score
    assert sorted(set(best)) <= list(range(k))
    # refined partitioning still yields the correct contraction
    ptn, path, _, _ = compute_solution(tn, best)
    ref = contract_network(
        network_to_otensors(tn),
        __import__("tnc_amd").Greedy().find_path(tn).replace_path(),
    )
    out = contract_network(network_to_otensors(ptn), path)
    np.testing.assert_allclose(out.data, ref.data, rtol=1e-10, atol=1e-12)


def test_memory_limit_rejects():
    tn = _net()
    k = 3
    part = find_partitioning(tn, k)
    model = NaivePartitioningModel(tn, k, memory_limit=1.0)  # 1 byte
    assert model.evaluate(part, np.random.default_rng(0)) == float("inf")
