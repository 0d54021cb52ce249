"""Single-GPU coverage of the pieces the RCCL fan-in backend composes:
engine result_dev -> tn_memcpy_dtod into a torch buffer (the send path),
tn_net_add_leaf_dev over torch memory (the recv path), pair contraction of
external device tensors (the fan-in merge)."""

import ctypes
import math

import numpy as np
import pytest

from oracle import contract_network, contract_tensors
from oracle.adapters import network_to_otensors

pytestmark = pytest.mark.gpu


def test_fanin_backend_interop():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from tnc_amd import Greedy, hiplib
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.dist import make_plan
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.tensor import CompositeTensor

    L = hiplib.lib()
    tn = random_circuit(12, 8, 0.5, 0.5, 21, ConnectivityLayout.EAGLE)
    plan = make_plan(tn, 2, trials=4)
    assert plan.nparts == 2

    # contract both partitions locally (as two "ranks" on one device)
    engines = {}
    for part in range(2):
        sub = plan.partitioned.tensors[part]
        inner = plan.path.nested.get(part)
        assert isinstance(sub, CompositeTensor) and inner is not None
        eng = ContractionEngine(sub, inner)
        eng.contract()
        engines[part] = eng

    # "send": copy partition-1's result into a torch buffer via the C ABI
    (x, y), = plan.path.toplevel[:1]
    ext_y = plan.externals[y]
    elems = max(1, int(math.prod(ext_y.bond_dims)))
    t = torch.empty((elems, 2), dtype=torch.float64, device="cuda:0")
    hiplib.check(
        L.tn_memcpy_dtod(t.data_ptr(), engines[y].result_dev(), elems * 16),
        "tn_memcpy_dtod",
    )
    # leg order of the device buffer must match the plan's simulated order
    legs_y, data_y = engines[y].result()
    assert legs_y == list(ext_y.legs)

    # "recv + merge": pair-contract local x with the torch-held tensor
    net = L.tn_net_create(0)
    assert net
    try:
        ext_x = plan.externals[x]
        ia = L.tn_net_add_leaf_dev(net, hiplib._u64arr(ext_x.legs),
                                   hiplib._u64arr(ext_x.bond_dims),
                                   len(ext_x.legs), engines[x].result_dev())
        ib = L.tn_net_add_leaf_dev(net, hiplib._u64arr(ext_y.legs),
                                   hiplib._u64arr(ext_y.bond_dims),
                                   len(ext_y.legs), t.data_ptr())
        assert ia == 0 and ib == 1
        hiplib.check(L.tn_net_contract(net, hiplib._u64arr([0, 1]), 1, None),
                     "pair contract")
        labels = (ctypes.c_uint64 * 64)()
        dims = (ctypes.c_uint64 * 64)()
        nd = ctypes.c_size_t()
        hiplib.check(L.tn_net_result_meta(net, labels, dims, ctypes.byref(nd)),
                     "meta")
        shape = tuple(dims[i] for i in range(nd.value))
        out = np.empty(shape, dtype=np.complex128)
        hiplib.check(
            L.tn_net_result_data(net, out.ctypes.data_as(ctypes.c_void_p)),
            "data")
        got_legs = [labels[i] for i in range(nd.value)]
    finally:
        L.tn_net_destroy(net)

    # oracle: contract partitions then merge
    ref_x = contract_network(
        network_to_otensors(plan.partitioned.tensors[x]), plan.path.nested[x])
    ref_y = contract_network(
        network_to_otensors(plan.partitioned.tensors[y]), plan.path.nested[y])
    ref = contract_tensors(ref_x, ref_y)
    assert got_legs == ref.legs
    np.testing.assert_allclose(out, ref.data, rtol=1e-10, atol=1e-12)
    for eng in engines.values():
        eng.close()
