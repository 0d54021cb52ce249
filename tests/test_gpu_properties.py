"""Full-size property tests on the benchmark fixture (SURVEY.md §8c /
DESIGN.md "Oracle"): the oracle cannot run the 36q network in test time, so
parity at BASELINE's full size rests on size-independent properties:

- linearity: the contraction is multilinear in every leaf, so scaling one
  leaf's data by alpha scales the (scalar) amplitude by exactly alpha;
- partition invariance: contracting the same network as two independently
  contracted partitions fanned in (the distributed execution shape,
  integration_tests.rs:26-86) reproduces the direct result.
"""

import ctypes

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _amplitude(tn, replace):
    from tnc_amd.executor import ContractionEngine

    eng = ContractionEngine(tn, replace)
    try:
        eng.contract()
        legs, data = eng.result()
        assert legs == []
        return complex(data)
    finally:
        eng.close()


def test_rqc36_linearity_full_size():
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.tensor import TensorData

    tn, rp, meta = load_fixture("rqc36")
    replace = ContractionPath.simple(rp)
    base = _amplitude(tn, replace)
    assert base != 0

    alpha = 0.5 - 0.25j
    leaf = tn.tensors[0]
    scaled = np.asarray(leaf.tensordata.into_data(), dtype=np.complex128) * alpha
    leaf.set_tensor_data(TensorData(TensorData.MATRIX, matrix=scaled))
    got = _amplitude(tn, replace)
    np.testing.assert_allclose(got / base, alpha, rtol=1e-10)


def test_rqc36_partitioned_equals_direct_full_size():
    """Two partitions contracted independently + pair-merged == the direct
    frozen-path amplitude (the exact shape the 8-GPU run executes, run
    sequentially on one device)."""
    from tnc_amd import hiplib
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.dist import make_plan
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.tensor import CompositeTensor

    tn, rp, meta = load_fixture("rqc36")
    direct = _amplitude(tn, ContractionPath.simple(rp))

    tn2, _, _ = load_fixture("rqc36")
    plan = make_plan(tn2, 2, trials=4, size_cap=2.0e9)
    engines = []
    try:
        for part in range(plan.nparts):
            sub = plan.partitioned.tensors[part]
            inner = plan.path.nested.get(part)
            assert isinstance(sub, CompositeTensor) and inner is not None
            eng = ContractionEngine(sub, inner)
            eng.contract()
            engines.append(eng)
        (x, y), = plan.path.toplevel
        L = hiplib.lib()
        net = L.tn_net_create(0)
        assert net
        try:
            for part, eng in ((x, engines[x]), (y, engines[y])):
                ext = plan.externals[part]
                idx = L.tn_net_add_leaf_dev(
                    net, hiplib._u64arr(ext.legs), hiplib._u64arr(ext.bond_dims),
                    len(ext.legs), eng.result_dev())
                assert idx >= 0
            hiplib.check(
                L.tn_net_contract(net, hiplib._u64arr([0, 1]), 1, None),
                "pair contract")
            out = np.empty((), dtype=np.complex128)
            hiplib.check(
                L.tn_net_result_data(net, out.ctypes.data_as(ctypes.c_void_p)),
                "result")
        finally:
            L.tn_net_destroy(net)
    finally:
        for eng in engines:
            eng.close()
    np.testing.assert_allclose(complex(out), direct, rtol=1e-10)
