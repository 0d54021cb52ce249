"""Greedy pathfinder pinned against the reference's expected SSA paths and
costs (contractionpath/paths/cotengrust.rs:229-307)."""

from tnc_amd import CompositeTensor, Greedy, LeafTensor, RandomGreedy
from tnc_amd.contraction_path import path, validate_path
from tnc_amd.paths import BasicContractionPathResult


def _setup_simple():
    bd = {0: 5, 1: 2, 2: 6, 3: 8, 4: 1, 5: 3, 6: 4}
    return CompositeTensor(
        [
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([0, 1, 3, 2], bd),
            LeafTensor.new_from_map([4, 5, 6], bd),
        ]
    )


def _setup_complex():
    bd = {0: 27, 1: 18, 2: 12, 3: 15, 4: 5, 5: 3, 6: 18, 7: 22, 8: 45, 9: 65,
          10: 5, 11: 17}
    return CompositeTensor(
        [
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([0, 1, 3, 2], bd),
            LeafTensor.new_from_map([4, 5, 6], bd),
            LeafTensor.new_from_map([6, 8, 9], bd),
            LeafTensor.new_from_map([10, 8, 9], bd),
            LeafTensor.new_from_map([5, 1, 0], bd),
        ]
    )


def test_greedy_simple():
    """cotengrust.rs:229-243."""
    result = Greedy().find_path(_setup_simple())
    assert result == BasicContractionPathResult(path((0, 1), (3, 2)), 600.0, 538.0)


def test_greedy_simple_inner():
    """cotengrust.rs:246-259."""
    bd = {0: 5, 1: 2, 2: 6, 3: 8, 4: 1, 5: 3, 6: 4}
    tn = CompositeTensor(
        [
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([0, 1, 5], bd),
            LeafTensor.new_from_map([1, 6], bd),
        ]
    )
    result = Greedy().find_path(tn)
    assert result == BasicContractionPathResult(
        path((0, 1), (2, 3), (4, 5)), 228.0, 121.0
    )


def test_greedy_simple_outer():
    """cotengrust.rs:262-275."""
    bd = {0: 3, 1: 2, 2: 2}
    tn = CompositeTensor(
        [
            LeafTensor.new_from_map([0], bd),
            LeafTensor.new_from_map([1], bd),
            LeafTensor.new_from_map([2], bd),
        ]
    )
    result = Greedy().find_path(tn)
    assert result == BasicContractionPathResult(path((2, 1), (0, 3)), 16.0, 19.0)


def test_greedy_complex_outer():
    """cotengrust.rs:278-291."""
    bd = {0: 5, 1: 4}
    tn = CompositeTensor(
        [
            LeafTensor.new_from_map([0], bd),
            LeafTensor.new_from_map([0], bd),
            LeafTensor.new_from_map([1], bd),
            LeafTensor.new_from_map([1], bd),
        ]
    )
    result = Greedy().find_path(tn)
    assert result == BasicContractionPathResult(
        path((0, 1), (2, 3), (5, 4)), 10.0, 11.0
    )


def test_greedy_complex():
    """cotengrust.rs:294-307."""
    result = Greedy().find_path(_setup_complex())
    assert result == BasicContractionPathResult(
        path((1, 5), (3, 4), (6, 0), (7, 2), (9, 8)), 529815.0, 89478.0
    )


def test_random_greedy_not_worse():
    tn = _setup_complex()
    g = Greedy().find_path(tn)
    rg = RandomGreedy(10).find_path(tn)
    assert rg.flops <= g.flops
    validate_path(rg.replace_path())


def test_greedy_nested_composites():
    """find_path recurses into composites (cotengrust.rs:120-137)."""
    bd = {0: 2, 1: 2, 2: 2, 3: 2, 4: 2}
    inner = CompositeTensor(
        [LeafTensor.new_from_map([0, 1], bd), LeafTensor.new_from_map([1, 2], bd)]
    )
    tn = CompositeTensor([inner, LeafTensor.new_from_map([2, 3], bd),
                          LeafTensor.new_from_map([3, 4], bd)])
    result = Greedy().find_path(tn)
    assert 0 in result.ssa_path.nested
    replace = result.replace_path()
    validate_path(replace)


def test_optimal_small():
    """Optimal finds a cost <= greedy on small nets and a valid path."""
    from tnc_amd import Optimal

    tn = _setup_complex()
    g = Greedy().find_path(tn)
    o = Optimal().find_path(tn)
    assert o.flops <= g.flops
    validate_path(o.replace_path())
    # greedy is already optimal on the pinned simple net
    tn2 = _setup_simple()
    assert Optimal().find_path(tn2).flops == 600.0


def test_optimal_disconnected():
    from tnc_amd import Optimal

    bd = {0: 3, 1: 2, 2: 2}
    tn = CompositeTensor(
        [
            LeafTensor.new_from_map([0], bd),
            LeafTensor.new_from_map([1], bd),
            LeafTensor.new_from_map([2], bd),
        ]
    )
    r = Optimal().find_path(tn)
    validate_path(r.replace_path())
    assert len(r.ssa_path.toplevel) == 2


def test_optimal_matches_oracle_result():
    import numpy as np

    from oracle import contract_network
    from oracle.adapters import network_to_otensors
    from tnc_amd import Optimal
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout

    tn = random_circuit(5, 4, 0.5, 0.5, 11, ConnectivityLayout.LINE
                        if hasattr(ConnectivityLayout, "LINE")
                        else ConnectivityLayout.Line(5))
    if len(tn.tensors) > 14:
        import pytest

        pytest.skip("too many tensors for optimal")
    ref = contract_network(network_to_otensors(tn),
                           Greedy().find_path(tn).replace_path())
    out = contract_network(network_to_otensors(tn),
                           Optimal().find_path(tn).replace_path())
    np.testing.assert_allclose(out.data, ref.data, rtol=1e-12)


def test_partition_search_beats_rg_on_rqc36():
    """The HyperOptimizer-substitute quality tier (config 3): PartitionSearch
    never loses to its own RandomGreedy baseline, stays valid and
    memory-capped, and the tier as frozen in the fixture (full ks/seeds,
    trials=64) reached an op cost far below what a quick RandomGreedy finds.
    (Since the greedy score formula was aligned with cotengra's convention,
    plain RandomGreedy itself closes most of the gap at small trial counts —
    the 3x headline win lives in the frozen fixture meta.)"""
    from tnc_amd import PartitionSearch, RandomGreedy
    from tnc_amd.fixtures import load_fixture

    tn, rp, meta = load_fixture("rqc36")
    rg = RandomGreedy(8, size_cap=6.0e9).find_path(tn)
    ps = PartitionSearch(ks=(2, 4), seeds=(0,), trials=8,
                         size_cap=6.0e9).find_path(tn)
    assert ps.flops <= rg.flops
    validate_path(ps.replace_path())
    assert ps.size <= 6.0e9
    # the frozen fixture path (found by the full-strength tier) is >=2x
    # cheaper than the quick RandomGreedy above
    assert meta["op_cost"] <= 0.5 * rg.flops


def test_partition_search_result_correct():
    import numpy as np

    from oracle import contract_network
    from oracle.adapters import network_to_otensors
    from tnc_amd import PartitionSearch
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout

    tn = random_circuit(14, 9, 0.5, 0.7, 9, ConnectivityLayout.EAGLE)
    ref = contract_network(network_to_otensors(tn),
                           Greedy().find_path(tn).replace_path())
    ps = PartitionSearch(ks=(2, 3), seeds=(0, 1), trials=4).find_path(tn)
    out = contract_network(network_to_otensors(tn), ps.replace_path())
    np.testing.assert_allclose(out.data, ref.data, rtol=1e-10, atol=1e-12)
