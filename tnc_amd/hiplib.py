"""ctypes binding of the tnc_hip C ABI (include/tnc_hip.h).

The product path fails loudly: importing this module without the built
library, or calling compute without a GPU, raises — there is no CPU
fallback anywhere in tnc_amd.
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libtnc_hip.so")

_lib = None


def lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                f"tnc_hip library not built: {_LIB_PATH} missing. "
                "Run `python -c 'import __graft_entry__; __graft_entry__.build()'`."
            )
        L = ctypes.CDLL(_LIB_PATH)
        u64p = ctypes.POINTER(ctypes.c_uint64)
        i64p = ctypes.POINTER(ctypes.c_int64)
        L.tn_last_error.restype = ctypes.c_char_p
        L.tn_device_count.restype = ctypes.c_int
        L.tn_set_device.argtypes = [ctypes.c_int]
        L.tn_einsum_c128.restype = ctypes.c_int
        L.tn_einsum_c128.argtypes = [
            u64p, u64p, ctypes.c_size_t,
            u64p, u64p, i64p, ctypes.c_void_p, ctypes.c_size_t,
            u64p, u64p, i64p, ctypes.c_void_p, ctypes.c_size_t,
            ctypes.c_void_p,
        ]
        L.tn_einsum_c128_dev.restype = ctypes.c_int
        L.tn_einsum_c128_dev.argtypes = [
            u64p, u64p, ctypes.c_size_t,
            u64p, u64p, i64p, ctypes.c_void_p, ctypes.c_size_t,
            u64p, u64p, i64p, ctypes.c_void_p, ctypes.c_size_t,
            ctypes.c_void_p, ctypes.c_void_p,
        ]
        L.tn_einsum_c64.restype = ctypes.c_int
        L.tn_einsum_c64.argtypes = L.tn_einsum_c128.argtypes
        L.tn_einsum_c64_dev.restype = ctypes.c_int
        L.tn_einsum_c64_dev.argtypes = L.tn_einsum_c128_dev.argtypes
        L.tn_net_create2.restype = ctypes.c_void_p
        L.tn_net_create2.argtypes = [ctypes.c_int, ctypes.c_int]
        L.tn_net_create.restype = ctypes.c_void_p
        L.tn_net_create.argtypes = [ctypes.c_int]
        L.tn_net_reserve.restype = ctypes.c_int
        L.tn_net_reserve.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
        L.tn_net_add_leaf.restype = ctypes.c_int64
        L.tn_net_add_leaf.argtypes = [
            ctypes.c_void_p, u64p, u64p, ctypes.c_size_t, ctypes.c_void_p,
        ]
        L.tn_net_add_leaf_dev.restype = ctypes.c_int64
        L.tn_net_add_leaf_dev.argtypes = [
            ctypes.c_void_p, u64p, u64p, ctypes.c_size_t, ctypes.c_void_p,
        ]
        L.tn_net_contract.restype = ctypes.c_int
        L.tn_net_contract.argtypes = [
            ctypes.c_void_p, u64p, ctypes.c_size_t, ctypes.POINTER(ctypes.c_double),
        ]
        L.tn_net_contract_profiled.restype = ctypes.c_int
        L.tn_net_contract_profiled.argtypes = [
            ctypes.c_void_p, u64p, ctypes.c_size_t,
            ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double),
            ctypes.POINTER(ctypes.c_int32), ctypes.POINTER(ctypes.c_double),
        ]
        L.tn_memcpy_dtod.restype = ctypes.c_int
        L.tn_memcpy_dtod.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint64,
        ]
        L.tn_memcpy_dtoh.restype = ctypes.c_int
        L.tn_memcpy_dtoh.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint64,
        ]
        L.tn_net_result_meta.restype = ctypes.c_int
        L.tn_net_result_meta.argtypes = [
            ctypes.c_void_p, u64p, u64p, ctypes.POINTER(ctypes.c_size_t),
        ]
        L.tn_net_result_data.restype = ctypes.c_int
        L.tn_net_result_data.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        L.tn_net_result_dev.restype = ctypes.c_void_p
        L.tn_net_result_dev.argtypes = [ctypes.c_void_p]
        L.tn_net_pool_bytes.restype = ctypes.c_int
        L.tn_net_pool_bytes.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint64),
            ctypes.POINTER(ctypes.c_uint64),
        ]
        L.tn_net_destroy.argtypes = [ctypes.c_void_p]
        _lib = L
    return _lib


def last_error() -> str:
    return lib().tn_last_error().decode()


def check(rc: int, what: str):
    if rc != 0:
        raise RuntimeError(f"{what} failed (rc={rc}): {last_error()}")


def device_count() -> int:
    return lib().tn_device_count()


def _u64arr(vals):
    return (ctypes.c_uint64 * len(vals))(*[int(v) for v in vals])


def _i64arr(vals):
    return (ctypes.c_int64 * len(vals))(*[int(v) for v in vals])


def _einsum_host(out_labels, a_labels, a, b_labels, b, out_shape, npdtype,
                 esize, fn_name):
    a = np.asarray(a, dtype=npdtype)
    b = np.asarray(b, dtype=npdtype)
    if out_shape is None:
        dimmap = {}
        for lab, d in zip(a_labels, a.shape):
            dimmap[lab] = d
        for lab, d in zip(b_labels, b.shape):
            dimmap[lab] = d
        out_shape = [dimmap[l] for l in out_labels]
    out = np.empty(tuple(out_shape), dtype=npdtype)
    a_str = [s // esize for s in a.strides]
    b_str = [s // esize for s in b.strides]
    rc = getattr(lib(), fn_name)(
        _u64arr(out_labels), _u64arr(out_shape), len(out_labels),
        _u64arr(a_labels), _u64arr(a.shape), _i64arr(a_str),
        a.ctypes.data_as(ctypes.c_void_p), a.ndim,
        _u64arr(b_labels), _u64arr(b.shape), _i64arr(b_str),
        b.ctypes.data_as(ctypes.c_void_p), b.ndim,
        out.ctypes.data_as(ctypes.c_void_p),
    )
    check(rc, fn_name)
    return out


def einsum_c128(out_labels, a_labels, a: np.ndarray, b_labels, b: np.ndarray,
                out_shape=None) -> np.ndarray:
    """Host-buffer einsum via the GPU (tn_einsum_c128): the direct
    tblis::tensor_mult parity entry point. Accepts non-contiguous views
    (strides forwarded in elements)."""
    return _einsum_host(out_labels, a_labels, a, b_labels, b, out_shape,
                        np.complex128, 16, "tn_einsum_c128")


def einsum_c64(out_labels, a_labels, a: np.ndarray, b_labels, b: np.ndarray,
               out_shape=None) -> np.ndarray:
    """complex64 einsum via the GPU (tn_einsum_c64, f32 MFMA path)."""
    return _einsum_host(out_labels, a_labels, a, b_labels, b, out_shape,
                        np.complex64, 8, "tn_einsum_c64")
