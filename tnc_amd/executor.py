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
from .tensor import CompositeTensor, LeafTensor


class StepInfo:
    __slots__ = ("i", "j", "m", "n", "k", "flops", "out_legs", "out_dims",
                 "packa", "packb", "pipe_p")

    def __init__(self, i, j, m, n, k, flops, out_legs, out_dims,
                 packa=False, packb=False, pipe_p=0):
        self.i, self.j = i, j
        self.m, self.n, self.k = m, n, k
        self.flops = flops
        self.out_legs = out_legs
        self.out_dims = out_dims
        self.packa = packa  # TTGT would permute A into [M..][K..] order
        self.packb = packb  # TTGT would permute B into [K..][N..] order
        self.pipe_p = pipe_p  # pack-pipeline window count (0 = serial pack)


def _pipeline_windows(m, n, k, packa, packb, kdims, esize):
    """Mirror of the C dispatch's pack-pipeline model (tnc_hip.hip
    einsum_dev_impl): number of K windows the pipelined TTGT would use,
    0 when the step packs serially. Keep in sync with the constants there
    (permute ~4 TB/s, stream ~6 TB/s, window tiles >= 256)."""
    if not packa:
        return 0
    m, n, k = int(m), int(n), int(k)
    if not (m >= 32 and n >= 32) or m % 128 or n % 64 or k % 16:
        return 0
    gemm_worthy = k >= 16 and m >= 128 and n >= 64
    if (k <= 64 or m < 16 or n < 16) and not gemm_worthy:
        return 0
    packbytes = m * k * esize + (k * n * esize if packb else 0)
    tiles_w = (m // 128) * ((n + 63) // 64)
    best_save, best_p = 0.0, 0
    p = 1
    for d in kdims:
        if p >= 16:
            break
        p *= int(d)
        if p < 2 or p > 16:
            continue
        if (k // p) % 16 or tiles_w < 256:
            continue
        save = (2.0 * packbytes / 4e12 * (1 - 1.0 / p)
                - 2.0 * p * (m * n * esize) / 6e12)
        if save > best_save:
            best_save, best_p = save, p
    # measured gate (see tnc_hip.hip): only strongly-favorable steps win
    if best_save <= 5e-3 or packbytes < 8.0 * m * n * esize:
        return 0
    return best_p


def plan_steps(leaves, steps):
    """Host-side metadata walk: per-step (M, N, K), metric flops
    ((8s-2)*o, contraction_cost.rs:26-32), output legs, and whether the
    TTGT path would need pack permutes (mirrors einsum_dev_impl's
    is_ready checks in tnc_amd/csrc/tnc_hip.hip)."""
    views = [LeafTensor(t.legs, t.bond_dims) for t in leaves]
    infos = []
    for i, j in steps:
        a, b = views[i], views[j]
        out = a ^ b
        shared = a & b
        m = (a - b).size()
        n = (b - a).size()
        k = shared.size()
        aset, bset = set(a.legs), set(b.legs)
        shared_a = [l for l in a.legs if l in bset]  # K legs in A order
        a_axes = [x for x, l in enumerate(a.legs) if l not in bset]
        a_axes += [a.legs.index(l) for l in shared_a]
        b_axes = [b.legs.index(l) for l in shared_a]
        b_axes += [x for x, l in enumerate(b.legs) if l not in aset]
        packa = a_axes != list(range(len(a.legs)))
        packb = b_axes != list(range(len(b.legs)))
        kdims = [a.bond_dims[a.legs.index(l)] for l in shared_a]
        # window count is esize-invariant (the save model scales linearly)
        pipe_p = _pipeline_windows(m, n, k, packa, packb, kdims, 16)
        infos.append(
            StepInfo(i, j, m, n, k, contract_cost_tensors(a, b), out.legs,
                     out.bond_dims, packa=packa, packb=packb, pipe_p=pipe_p)
        )
        views[i] = out
        views[j] = None
    return infos


def _pack_ws_bytes(info, esize):
    """Pack-permute workspace of a TTGT step (0 for dot/gather routes)."""
    gemm_worthy = info.k >= 16 and info.m >= 128 and info.n >= 64
    if (info.k <= 64 or info.m < 16 or info.n < 16) and not gemm_worthy:
        return 0
    ws = 0
    if info.packa:
        ws += info.m * info.k * esize
    if info.packb:
        ws += info.k * info.n * esize
    return ws


def _step_ws_bytes(info, esize):
    """Workspace the C dispatch will allocate for one step (mirrors
    einsum_dev_impl: dot partials, nothing for gather kernels, pack
    buffers for TTGT). Keep in sync with TN_SMALLK=64, MF_T=128,
    MF_TN=64 in tnc_amd/csrc/tnc_hip.hip."""
    if info.m == 1 and info.n == 1 and info.k > 64:
        return 1 << 25  # dot partial buffer (<= 2^21 blocks * 16 B)
    # pipelined steps additionally hold pipe_p split-K-style output slices
    return (_pack_ws_bytes(info, esize)
            + info.pipe_p * int(info.m) * int(info.n) * esize)


def arena_bytes(leaves, steps, infos, esize=16):
    """Peak device-arena demand: live intermediates + this step's output and
    the workspaces its dispatch actually uses, walked over the plan. The
    pack-overlap executor allocates the NEXT step's pack buffers one step
    early (prepack on stream2), so each step's demand includes them.
    Padded 15% for fragmentation + split-K slack."""
    live = {}  # slot -> bytes (intermediates only; leaves live outside)
    peak = 0
    for s, info in enumerate(infos):
        out_b = info.m * info.n * esize
        nxt = _pack_ws_bytes(infos[s + 1], esize) if s + 1 < len(infos) else 0
        demand = (sum(live.values()) + out_b + _step_ws_bytes(info, esize)
                  + nxt)
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
        # launch-bound walks (many tiny steps) only replay as a hipGraph
        # when every workspace comes from the arena — reserve a floor so
        # replay engages (rqc24: ~2.4 ms/contraction unreserved vs
        # ~0.6 ms replayed)
        if len(steps) >= 64:
            reserve = max(reserve, 256 * 1024 * 1024)
        # arena is an optimization: if the reservation fails (tiny GPUs,
        # fragmented memory), every block falls back to plain hipMalloc
        # (the stream-ordered pool is avoided — see ws_alloc in tnc_hip.hip)
        if reserve > 64 * 1024 * 1024:
            rc = L.tn_net_reserve(self.net, reserve)
            if rc != 0:
                import warnings

                warnings.warn(
                    f"arena reservation of {reserve} bytes failed "
                    f"({hiplib.last_error()}); falling back to per-block "
                    "hipMalloc"
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
