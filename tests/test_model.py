"""Host tensor model: leg algebra, paths, cost model — pinned against the
reference's own unit tests (exact expected values)."""

import pytest

from tnc_amd import (
    CompositeTensor,
    ContractionPath,
    LeafTensor,
    TensorData,
    communication_path_cost,
    contract_cost_tensors,
    contract_op_cost_tensors,
    contract_path_cost,
    ssa_replace_ordering,
    validate_path,
)
from tnc_amd.contraction_path import flatten_network, path, ssa_ordering
from tnc_amd.cost import contract_size_tensors, contract_size_tensors_bytes


def test_leg_algebra():
    """tensor.rs doctests (:615-725)."""
    bd = {1: 2, 2: 4, 3: 6, 4: 3, 5: 9}
    t1 = LeafTensor.new_from_map([1, 2, 3], bd)
    t2 = LeafTensor.new_from_map([4, 2, 5], bd)
    assert (t1 - t2).legs == [1, 3] and (t1 - t2).bond_dims == [2, 6]
    assert (t1 | t2).legs == [1, 2, 3, 4, 5]
    assert (t1 | t2).bond_dims == [2, 4, 6, 3, 9]
    assert (t1 & t2).legs == [2] and (t1 & t2).bond_dims == [4]
    assert (t1 ^ t2).legs == [1, 3, 4, 5]
    assert (t1 ^ t2).bond_dims == [2, 6, 3, 9]


def test_external_tensor():
    """tensor.rs:893-917."""
    bd = {2: 2, 3: 4, 4: 6, 5: 8, 6: 10, 7: 12, 8: 14, 9: 16}
    t12 = CompositeTensor(
        [LeafTensor.new_from_map([2, 3, 4], bd), LeafTensor.new_from_map([2, 3, 5], bd)]
    )
    t34 = CompositeTensor(
        [LeafTensor.new_from_map([6, 7, 8], bd), LeafTensor.new_from_map([6, 8, 9], bd)]
    )
    ext = CompositeTensor([t12, t34]).external_tensor()
    assert ext.legs == [4, 5, 7, 9]
    assert ext.bond_dims == [6, 8, 12, 16]


def test_ssa_ordering():
    """contractionpath.rs:268-283."""
    raw = [(0, 3, 15), (1, 2, 44), (6, 4, 8), (5, 15, 22), (8, 44, 12), (12, 22, 99)]
    assert ssa_ordering(raw, 7) == path((0, 3), (1, 2), (6, 4), (5, 7), (9, 8), (11, 10))


def test_ssa_replace_ordering():
    """contractionpath.rs:286-294."""
    p = path((0, 3), (1, 2), (6, 4), (5, 7), (9, 8), (11, 10))
    assert ssa_replace_ordering(p) == path((0, 3), (1, 2), (6, 4), (5, 0), (6, 1), (6, 5))


def test_ssa_replace_ordering_nested():
    """contractionpath.rs:297-328."""
    p = path(
        (0, 3), (1, 2), (6, 4), (5, 7), (9, 8), (11, 10),
        nested={1: path((2, 1), (0, 3)), 6: path((0, 2), (1, 3), (4, 5))},
    )
    expect = path(
        (0, 3), (1, 2), (6, 4), (5, 0), (6, 1), (6, 5),
        nested={1: path((2, 1), (0, 2)), 6: path((0, 2), (1, 3), (0, 1))},
    )
    assert ssa_replace_ordering(p) == expect


def test_validate_path():
    """paths.rs:108-115."""
    with pytest.raises(AssertionError):
        validate_path(path((0, 1), (1, 2)))
    validate_path(path((0, 1), (0, 2)))


def _setup_simple():
    bd = {0: 5, 1: 2, 2: 6, 3: 8, 4: 1, 5: 3, 6: 4}
    return CompositeTensor(
        [
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([0, 1, 3, 2], bd),
            LeafTensor.new_from_map([4, 5, 6], bd),
        ]
    )


def _setup_complex_nested():
    bd = {0: 5, 1: 2, 2: 6, 3: 8, 4: 1, 5: 3, 6: 4, 7: 3, 8: 2, 9: 2}
    t1 = CompositeTensor(
        [
            LeafTensor.new_from_map([4, 3, 2], bd),
            LeafTensor.new_from_map([0, 1, 3, 2], bd),
            LeafTensor.new_from_map([4, 5, 6], bd),
        ]
    )
    t2 = CompositeTensor(
        [
            LeafTensor.new_from_map([5, 6, 8], bd),
            LeafTensor.new_from_map([7, 8, 9], bd),
        ]
    )
    return CompositeTensor([t1, t2])


def test_cost_doctest_values():
    """contraction_cost.rs doctests (:16-91)."""
    bd = {0: 5, 1: 7, 2: 9, 3: 11, 4: 13}
    t1 = LeafTensor.new_from_map([0, 1, 2], bd)
    t2 = LeafTensor.new_from_map([2, 3, 4], bd)
    assert contract_cost_tensors(t1, t2) == 350350.0
    assert contract_op_cost_tensors(t1, t2) == 45045.0
    assert contract_size_tensors(t1, t2) == 6607.0
    assert contract_size_tensors_bytes(t1, t2) == 6607.0 * 16


def test_contract_path_cost():
    """contraction_cost.rs:324-356."""
    tn = _setup_simple()
    assert contract_path_cost(tn.tensors, path((0, 1), (0, 2)), False) == (4540.0, 538.0)
    assert contract_path_cost(tn.tensors, path((0, 2), (0, 1)), False) == (49296.0, 1176.0)
    assert contract_path_cost(tn.tensors, path((0, 1), (0, 2)), True) == (600.0, 538.0)
    assert contract_path_cost(tn.tensors, path((0, 2), (0, 1)), True) == (6336.0, 1176.0)


def test_contract_path_cost_nested():
    """contraction_cost.rs:336-345, 359-368."""
    tn = _setup_complex_nested()
    p = path((0, 1), nested={0: path((0, 1), (0, 2)), 1: path((0, 1))})
    assert contract_path_cost(tn.tensors, p, False) == (11188.0, 538.0)
    assert contract_path_cost(tn.tensors, p, True) == (1464.0, 538.0)


def _setup_parallel():
    bd = {0: 5, 1: 2, 2: 6, 3: 8, 4: 1, 5: 3, 6: 4}
    return [
        LeafTensor.new_from_map([4, 3, 2], bd),
        LeafTensor.new_from_map([0, 1, 3, 2], bd),
        LeafTensor.new_from_map([4, 5, 6], bd),
        LeafTensor.new_from_map([5, 6], bd),
    ]


def test_communication_path_cost():
    """contraction_cost.rs:371-416."""
    ts = _setup_parallel()
    assert communication_path_cost(ts, [(0, 1), (2, 3), (0, 2)], True, True) == (490.0, 538.0)
    assert communication_path_cost(ts, [(0, 1), (2, 3), (0, 1)], False, True) == (7564.0, 538.0)
    tc = [20.0, 30.0, 80.0, 10.0]
    assert communication_path_cost(ts, [(0, 1), (2, 3), (0, 2)], True, True, tc) == (520.0, 538.0)
    assert communication_path_cost(ts, [(0, 1), (2, 3), (0, 1)], False, True, tc) == (7594.0, 538.0)


def test_flatten_network_nested():
    """flatten matches the recursive walk's contraction set and final slot."""
    tn = _setup_complex_nested()
    p = path((0, 1), nested={0: path((0, 1), (0, 2)), 1: path((0, 1))})
    leaves, steps, final = flatten_network(tn, p)
    assert len(leaves) == 5
    assert steps == [(0, 1), (0, 2), (3, 4), (0, 3)]
    assert final == 0
