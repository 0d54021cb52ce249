"""Full-size VALUE parity on the benchmark fixtures.

The rqc36 headline fixture is contracted IN FULL by the oracle on the box's
host cores (peak live ~46 GB; RAM-guarded skip) and compared to the GPU
amplitude at the north star's 1e-10 relative bar — value agreement at the
exact benchmark workload, not just self-consistency properties.

syc49 (config 5, c64) gets the same property suite rqc36 already has:
full-size linearity and partition-invariance (integration_tests.rs:26-86
semantics via the tree-cut plan).
"""

import ctypes

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _amplitude(tn, replace, dtype="c128"):
    from tnc_amd.executor import ContractionEngine

    eng = ContractionEngine(tn, replace, dtype=dtype)
    try:
        eng.contract()
        legs, data = eng.result()
        assert legs == []
        return complex(data)
    finally:
        eng.close()


def test_rqc36_full_value_vs_oracle():
    """The flagship parity check: GPU amplitude of the exact bench fixture ==
    oracle's full-network contraction, 1e-10 relative (north star bar)."""
    from bench import _mem_available_bytes, oracle_peak_bytes
    from oracle import contract_network
    from oracle.adapters import network_to_otensors
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.fixtures import load_fixture

    tn, rp, meta = load_fixture("rqc36")
    avail = _mem_available_bytes()
    need = oracle_peak_bytes(tn, rp) * 1.7
    if avail is not None and need > avail:
        pytest.skip(f"host RAM too small for the full oracle walk "
                    f"(need ~{need/1e9:.0f} GB, available {avail/1e9:.0f} GB)")

    gpu = _amplitude(tn, ContractionPath.simple(rp))
    ref = contract_network(network_to_otensors(tn), rp)
    assert ref.legs == []
    np.testing.assert_allclose(gpu, complex(ref.data), rtol=1e-10)


def test_syc49_linearity_full_size():
    """Scaling one leaf scales the amplitude by exactly alpha (multilinearity)
    at the full config-5 size, c64 path."""
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.tensor import TensorData

    tn, rp, meta = load_fixture("syc49")
    assert meta.get("dtype") == "c64"
    replace = ContractionPath.simple(rp)
    base = _amplitude(tn, replace, dtype="c64")
    assert base != 0

    alpha = 0.5 - 0.25j
    leaf = tn.tensors[0]
    scaled = np.asarray(leaf.tensordata.into_data(), dtype=np.complex128) * alpha
    leaf.set_tensor_data(TensorData(TensorData.MATRIX, matrix=scaled))
    got = _amplitude(tn, replace, dtype="c64")
    # ratio of two c64 contractions: the shared rounding largely cancels,
    # but keep a c64-honest bar
    np.testing.assert_allclose(got / base, alpha, rtol=5e-4)


def test_syc49_partitioned_equals_direct_full_size():
    """Tree-cut 2-way partition of the frozen syc49 path, contracted as two
    independent engines + a pair merge of the device-resident externals,
    equals the direct amplitude (the distributed execution shape on one
    device, c64)."""
    from tnc_amd import hiplib
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.dist import make_tree_plan
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.tensor import CompositeTensor

    tn, rp, meta = load_fixture("syc49")
    direct = _amplitude(tn, ContractionPath.simple(rp), dtype="c64")

    tn2, _, _ = load_fixture("syc49")
    plan = make_tree_plan(tn2, rp, 2)
    engines = []
    try:
        for part in range(plan.nparts):
            sub = plan.partitioned.tensors[part]
            inner = plan.path.nested.get(part)
            assert isinstance(sub, CompositeTensor) and inner is not None
            eng = ContractionEngine(sub, inner, dtype="c64")
            eng.contract()
            engines.append(eng)
        (x, y), = plan.path.toplevel
        L = hiplib.lib()
        net = L.tn_net_create2(0, 1)  # dtype 1 = c64
        assert net
        try:
            for part, eng in ((x, engines[x]), (y, engines[y])):
                ext = plan.externals[part]
                idx = L.tn_net_add_leaf_dev(
                    net, hiplib._u64arr(ext.legs),
                    hiplib._u64arr(ext.bond_dims), len(ext.legs),
                    eng.result_dev())
                assert idx >= 0
            hiplib.check(
                L.tn_net_contract(net, hiplib._u64arr([0, 1]), 1, None),
                "pair contract")
            out = np.empty((), dtype=np.complex64)
            hiplib.check(
                L.tn_net_result_data(net, out.ctypes.data_as(ctypes.c_void_p)),
                "result")
        finally:
            L.tn_net_destroy(net)
    finally:
        for eng in engines:
            eng.close()
    # two c64 contractions in different orders over ~3.7e14 flops
    np.testing.assert_allclose(complex(out), direct, rtol=2e-3)
