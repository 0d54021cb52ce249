"""C-ABI surface: the built library loads and exports every symbol declared
in include/tnc_hip.h (runs on CPU; no compute)."""

import ctypes
import os
import re

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(ROOT, "include", "tnc_hip.h")
LIB = os.path.join(ROOT, "tnc_amd", "libtnc_hip.so")


def declared_symbols():
    with open(HEADER) as f:
        text = f.read()
    # function declarations: "<ret> tn_xxx(" at top level
    return sorted(set(re.findall(r"\b(tn_[a-z0-9_]+)\s*\(", text)))


@pytest.mark.skipif(not os.path.exists(LIB), reason="library not built")
def test_exports_match_header():
    lib = ctypes.CDLL(LIB)
    syms = declared_symbols()
    assert len(syms) >= 14
    for s in syms:
        assert hasattr(lib, s), f"missing export: {s}"


@pytest.mark.skipif(not os.path.exists(LIB), reason="library not built")
def test_no_gpu_fails_loudly():
    """Without a GPU the compute entry points must error, never fall back."""
    lib = ctypes.CDLL(LIB)
    lib.tn_device_count.restype = ctypes.c_int
    if lib.tn_device_count() > 0:
        pytest.skip("GPU present")
    lib.tn_net_create.restype = ctypes.c_void_p
    assert lib.tn_net_create(0) in (None, 0)
    lib.tn_last_error.restype = ctypes.c_char_p
    assert b"no AMD GPU" in lib.tn_last_error()
