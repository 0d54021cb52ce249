"""Parity of the env-gated alternate dispatch paths.

The pack-pipeline (TN_PIPELINE_FORCE) and the opt-in gather-staged GEMM
(TN_GATHER_GEMM) read their env switches once at library load, so each
case runs in a subprocess with the env set and asserts the oracle
comparison inside.
"""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

PIPELINE_CASE = r"""
import ctypes
import numpy as np
import oracle
from tnc_amd import hiplib
from tnc_amd.executor import ContractionEngine
from tnc_amd.tensor import CompositeTensor, LeafTensor, TensorData

rng = np.random.default_rng(7)
# one TTGT step: M = 2048 (legs 8x16x16), K = 256 (legs 4x4x4x4),
# N = 1024 (legs 16x64); A's K legs interleaved -> packa; window split
# over the leading K leg (dim 4). tiles_w = (2048/128)*(1024/64) = 256.
m_legs, m_dims = [1, 2, 3], [8, 16, 16]
k_legs, k_dims = [10, 11, 12, 13], [4, 4, 4, 4]
n_legs, n_dims = [20, 21], [16, 64]
a_legs = [10, 1, 11, 2, 12, 3, 13]
a_dims = [4, 8, 4, 16, 4, 16, 4]
# B interleaved too -> packb: exercises the window chunk buffers
b_legs = [10, 20, 11, 12, 13, 21]
b_dims = [4, 16, 4, 4, 4, 64]
a = (rng.standard_normal(a_dims) + 1j * rng.standard_normal(a_dims))
b = (rng.standard_normal(b_dims) + 1j * rng.standard_normal(b_dims))
ta = LeafTensor(a_legs, a_dims); ta.set_tensor_data(TensorData(TensorData.MATRIX, matrix=a))
tb = LeafTensor(b_legs, b_dims); tb.set_tensor_data(TensorData(TensorData.MATRIX, matrix=b))
from tnc_amd.contraction_path import ContractionPath
eng = ContractionEngine(CompositeTensor([ta, tb]), ContractionPath.simple([(0, 1)]))
# pipeline needs an arena (stream2 packs allocate from it)
hiplib.check(hiplib.lib().tn_net_reserve(eng.net, 1024 * 1024 * 1024), "reserve")
eng.contract()
legs, data = eng.result()
eng.close()
ref = oracle.contract_ndarrays(legs, a_legs, a, b_legs, b)
np.testing.assert_allclose(data, ref, rtol=1e-12, atol=1e-10)
print("PIPELINE_CASE_OK")
"""

GATHER_CASE = r"""
import numpy as np
import oracle
from tnc_amd import hiplib

rng = np.random.default_rng(9)
# pure pow2 TTGT shape with packa+packb: M = 256 (2 legs), N = 64,
# K = 64 (legs 4x4x4, interleaved in A)
a_legs = [10, 1, 11, 2, 12]
a_dims = [4, 16, 4, 16, 4]
b_legs = [12, 20, 10, 11]
b_dims = [4, 64, 4, 4]
a = (rng.standard_normal(a_dims) + 1j * rng.standard_normal(a_dims))
b = (rng.standard_normal(b_dims) + 1j * rng.standard_normal(b_dims))
from oracle.core import symmetric_difference
out_labels, _ = symmetric_difference(a_legs, a_dims, b_legs, b_dims)
ref = oracle.contract_ndarrays(out_labels, a_legs, a, b_legs, b)
got = hiplib.einsum_c128(out_labels, a_legs, a, b_legs, b)
np.testing.assert_allclose(got, ref, rtol=1e-12, atol=1e-10)
print("GATHER_CASE_OK")
"""


def _run_case(code, env_extra, expect):
    env = dict(os.environ)
    env.update(env_extra)
    proc = subprocess.run([sys.executable, "-c", code], cwd=ROOT, env=env,
                          capture_output=True, text=True, timeout=240)
    assert proc.returncode == 0, (proc.stdout, proc.stderr)
    assert expect in proc.stdout, (proc.stdout, proc.stderr)


def test_forced_pipeline_parity():
    """The K-window pack pipeline (stream2 permutes + split-K-slice GEMMs
    + reduce) reproduces the oracle on a small shape the profitability
    gate would normally reject."""
    _run_case(PIPELINE_CASE, {"TN_PIPELINE_FORCE": "1"}, "PIPELINE_CASE_OK")


def test_forced_pipeline_matches_default():
    """Same case through the default (serial-pack) dispatch — guards the
    fixture itself."""
    _run_case(PIPELINE_CASE, {"TN_NO_PIPELINE": "1"}, "PIPELINE_CASE_OK")


def test_gather_gemm_parity():
    """The opt-in gather-staged GEMM (TN_GATHER_GEMM=1) reproduces the
    oracle through the host einsum entry."""
    _run_case(GATHER_CASE, {"TN_GATHER_GEMM": "1"}, "GATHER_CASE_OK")
