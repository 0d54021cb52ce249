"""GPU contraction engine: the contract_tensor_network drop-in
(contraction.rs:35) running entirely on the MI355X.

Wraps the tn_net_* C ABI: leaves are uploaded once, the (flattened)
replace-left path walks on-device, intermediates never touch the host.
"""

from __future__ import annotations

import ctypes

import numpy as np

from . import hiplib
from .contraction_path import ContractionPath, flatten_network
from .cost import contract_cost_tensors
from .tensor import CompositeTensor, LeafTensor, TensorData


class StepInfo:
    __slots__ = ("i", "j", "m", "n", "k", "flops", "out_legs", "out_dims")

    def __init__(self, i, j, m, n, k, flops, out_legs, out_dims):
        self.i, self.j = i, j
        self.m, self.n, self.k = m, n, k
        self.flops = flops
        self.out_legs = out_legs
        self.out_dims = out_dims


def plan_steps(leaves, steps):
    """Host-side metadata walk: per-step (M, N, K), metric flops
    ((8s-2)*o, contraction_cost.rs:26-32) and output legs."""
    views = [LeafTensor(t.legs, t.bond_dims) for t in leaves]
    infos = []
    for i, j in steps:
        a, b = views[i], views[j]
        out = a ^ b
        shared = a & b
        m = (a - b).size()
        n = (b - a).size()
        k = shared.size()
        infos.append(
            StepInfo(i, j, m, n, k, contract_cost_tensors(a, b), out.legs, out.bond_dims)
        )
        views[i] = out
        views[j] = None
    return infos


def arena_bytes(leaves, steps, infos, esize=16):
    """Peak device-arena demand: live intermediates + this step's output and
    worst-case packing workspaces (A' + B' + unpacked C), walked over the
    plan. `esize` bytes per element; padded 15% for fragmentation."""
    live = {}  # slot -> bytes (intermediates only; leaves live outside)
    peak = 0
    for info in infos:
        out_b = info.m * info.n * esize
        extra = (info.m * info.k + info.k * info.n) * esize + out_b
        demand = sum(live.values()) + out_b + extra
        peak = max(peak, demand)
        live.pop(info.j, None)
        live[info.i] = out_b
    return int(peak * 1.15) + (1 << 20)


class ContractionEngine:
    """Device-resident executor for one (flattened) network."""

    def __init__(self, tn: CompositeTensor, replace_path: ContractionPath,
                 device=0, dtype="c128"):
        if hiplib.device_count() == 0:
            raise RuntimeError("no AMD GPU present — tnc_amd has no CPU fallback")
        assert dtype in ("c128", "c64")
        self.dtype = dtype
        self.npdtype = np.complex128 if dtype == "c128" else np.complex64
        self.esize = 16 if dtype == "c128" else 8
        leaves, steps, final = flatten_network(tn, replace_path)
        self.leaves = leaves
        self.steps = steps
        self.final = final
        self.infos = plan_steps(leaves, steps)
        self.total_flops = sum(s.flops for s in self.infos)
        L = hiplib.lib()
        hiplib.check(L.tn_set_device(device), "tn_set_device")
        self.net = L.tn_net_create2(device, 0 if dtype == "c128" else 1)
        if not self.net:
            raise RuntimeError(f"tn_net_create failed: {hiplib.last_error()}")
        reserve = arena_bytes(leaves, steps, self.infos, self.esize)
        # arena is an optimization: if the reservation fails (tiny GPUs,
        # fragmented memory), the executor falls back to hipMallocAsync
        if reserve > 64 * 1024 * 1024:
            rc = L.tn_net_reserve(self.net, reserve)
            if rc != 0:
                import warnings

                warnings.warn(
                    f"arena reservation of {reserve} bytes failed "
                    f"({hiplib.last_error()}); falling back to async allocs"
                )
        for t in leaves:
            data = np.ascontiguousarray(t.tensordata.into_data(), dtype=self.npdtype)
            assert list(data.shape) == list(t.bond_dims), (data.shape, t.bond_dims)
            idx = L.tn_net_add_leaf(
                self.net,
                hiplib._u64arr(t.legs),
                hiplib._u64arr(t.bond_dims),
                len(t.legs),
                data.ctypes.data_as(ctypes.c_void_p),
            )
            if idx < 0:
                raise RuntimeError(f"tn_net_add_leaf failed: {hiplib.last_error()}")
        self._pairs = hiplib._u64arr([x for p in steps for x in p])

    def contract(self) -> float:
        """One full contraction; returns device wall ms."""
        ms = ctypes.c_double()
        hiplib.check(
            hiplib.lib().tn_net_contract(
                self.net, self._pairs, len(self.steps), ctypes.byref(ms)
            ),
            "tn_net_contract",
        )
        return ms.value

    def contract_profiled(self):
        """Contraction with per-step HIP-event timings.
        Returns (elapsed_ms, step_ms[], gemm_ms[], kind[])."""
        n = len(self.steps)
        step_ms = (ctypes.c_double * n)()
        gemm_ms = (ctypes.c_double * n)()
        kind = (ctypes.c_int32 * n)()
        ms = ctypes.c_double()
        hiplib.check(
            hiplib.lib().tn_net_contract_profiled(
                self.net, self._pairs, n, step_ms, gemm_ms, kind, ctypes.byref(ms)
            ),
            "tn_net_contract_profiled",
        )
        return ms.value, list(step_ms), list(gemm_ms), list(kind)

    def result(self) -> "tuple[list, np.ndarray]":
        L = hiplib.lib()
        labels = (ctypes.c_uint64 * 64)()
        dims = (ctypes.c_uint64 * 64)()
        nd = ctypes.c_size_t()
        hiplib.check(
            L.tn_net_result_meta(self.net, labels, dims, ctypes.byref(nd)),
            "tn_net_result_meta",
        )
        shape = tuple(dims[i] for i in range(nd.value))
        legs = [labels[i] for i in range(nd.value)]
        out = np.empty(shape, dtype=self.npdtype)
        hiplib.check(
            L.tn_net_result_data(self.net, out.ctypes.data_as(ctypes.c_void_p)),
            "tn_net_result_data",
        )
        return legs, out

    def result_dev(self) -> int:
        """Device pointer of the final tensor (valid until next contract)."""
        ptr = hiplib.lib().tn_net_result_dev(self.net)
        if not ptr:
            raise RuntimeError("no result available")
        return ptr

    def close(self):
        if getattr(self, "net", None):
            hiplib.lib().tn_net_destroy(self.net)
            self.net = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def contract_tensor_network_gpu(tn: CompositeTensor, replace_path, device=0):
    """One-shot: contract and return (legs, ndarray)."""
    eng = ContractionEngine(tn, replace_path, device)
    try:
        eng.contract()
        return eng.result()
    finally:
        eng.close()
