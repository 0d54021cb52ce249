"""PEPS builder structure pinned to the reference's own unit tests
(builders/peps.rs:470-600 expected leg lists) plus closure and a
data-filled contraction through the oracle."""

from collections import Counter

import numpy as np

from oracle import contract_network
from oracle.adapters import network_to_otensors
from tnc_amd import Greedy
from tnc_amd.builders import peps
from tnc_amd.tensor import TensorData


def test_peps_init_layer_matches_reference():
    tn = peps(3, 3, 4, 10, 1)
    legs = [list(t.legs) for t in tn.tensors]
    assert legs[:9] == [
        [0, 9, 15], [1, 9, 10, 16], [2, 10, 17],
        [3, 11, 15, 18], [4, 11, 12, 16, 19], [5, 12, 17, 20],
        [6, 13, 18], [7, 13, 14, 19], [8, 14, 20],
    ]
    dims = [list(t.bond_dims) for t in tn.tensors]
    assert dims[0] == [4, 10, 10]
    assert dims[4] == [4, 10, 10, 10, 10]


def test_peps_pepo_layer_matches_reference():
    tn = peps(3, 3, 4, 10, 1)
    legs = [list(t.legs) for t in tn.tensors]
    assert legs[9:18] == [
        [0, 21, 30, 36], [1, 22, 30, 31, 37], [2, 23, 31, 38],
        [3, 24, 32, 36, 39], [4, 25, 32, 33, 37, 40], [5, 26, 33, 38, 41],
        [6, 27, 34, 39], [7, 28, 34, 35, 40], [8, 29, 35, 41],
    ]


def test_peps_closed_network():
    for L, D, layers in ((2, 2, 0), (3, 2, 1), (3, 3, 2)):
        tn = peps(L, D, 2, 3, layers)
        assert len(tn.tensors) == L * D * (layers + 2)
        c = Counter(l for t in tn.tensors for l in t.legs)
        assert set(c.values()) == {2}  # closed: every edge shared exactly 2x


def test_peps_contracts_to_scalar():
    rng = np.random.default_rng(5)
    tn = peps(2, 3, 2, 2, 1)
    for t in tn.tensors:
        data = (rng.standard_normal(t.shape) +
                1j * rng.standard_normal(t.shape))
        t.set_tensor_data(TensorData(TensorData.MATRIX, matrix=data))
    replace = Greedy().find_path(tn).replace_path()
    out = contract_network(network_to_otensors(tn), replace)
    assert out.legs == []
    assert np.isfinite(out.data).all()
