"""GPU backend for the partition-fan-in distributed contraction
(communication.rs:199-249 semantics; plan/walk in tnc_amd/dist.py).

Each rank contracts its (tree-cut) partition with a ContractionEngine;
open-leg intermediates travel as torch tensors (float views of c128/c64
device buffers over RCCL, or host round-trips under gloo); the receiving
rank pair-merges {local, received} through a transient tn_net with
device-resident leaves (tn_net_add_leaf_dev). Rank 0 ends up with the
final tensor (communication.rs:236-247).
"""

from __future__ import annotations

import ctypes

import numpy as np

from . import hiplib
from .dist import DistPlan, run_fanin
from .executor import ContractionEngine
from .tensor import CompositeTensor


class _Handle:
    """A device-resident tensor plus whatever owns its memory."""

    __slots__ = ("ptr", "elems", "owner")

    def __init__(self, ptr, elems, owner):
        self.ptr = ptr
        self.elems = elems
        self.owner = owner  # engine / tn_net wrapper / torch tensor


class _PairNet:
    """Transient tn_net holding one pair contraction's result."""

    def __init__(self, device, dtype_code):
        L = hiplib.lib()
        self.net = L.tn_net_create2(device, dtype_code)
        if not self.net:
            raise RuntimeError(f"tn_net_create2: {hiplib.last_error()}")

    def close(self):
        if getattr(self, "net", None):
            hiplib.lib().tn_net_destroy(self.net)
            self.net = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def run_fanin_gpu(plan: DistPlan, rank: int, world: int, dist_t, torch,
                  device, dev_id=0, dtype="c128", backend="nccl"):
    """Execute the plan: local contraction + fan-in. Returns a _Handle on
    rank 0 (final tensor, device-resident) and None elsewhere. Caller
    must keep the returned handle alive while reading it."""
    esize = 16 if dtype == "c128" else 8
    torch_view = torch.float64 if dtype == "c128" else torch.float32
    dtype_code = 0 if dtype == "c128" else 1
    L = hiplib.lib()

    my_part = None
    for p, r in plan.part_rank.items():
        if r == rank:
            my_part = p

    local = None
    if my_part is not None:
        sub = plan.partitioned.tensors[my_part]
        inner = plan.path.nested.get(my_part)
        if isinstance(sub, CompositeTensor) and inner is not None and \
                inner.toplevel:
            eng = ContractionEngine(sub, inner, device=dev_id, dtype=dtype)
            eng.contract()
            ext = plan.externals[my_part]
            local = _Handle(eng.result_dev(), int(ext.size()), eng)
        else:
            # single-leaf partition: upload the leaf through a pair net
            # holder (leaf data persists inside the net)
            leaf = sub.tensors[0] if isinstance(sub, CompositeTensor) else sub
            holder = _PairNet(dev_id, dtype_code)
            data = np.ascontiguousarray(
                leaf.tensordata.into_data(),
                dtype=np.complex128 if dtype == "c128" else np.complex64)
            idx = L.tn_net_add_leaf(
                holder.net, hiplib._u64arr(leaf.legs),
                hiplib._u64arr(leaf.bond_dims), len(leaf.legs),
                data.ctypes.data_as(ctypes.c_void_p))
            assert idx >= 0, hiplib.last_error()
            # contract the trivial single-leaf path to expose result_dev
            hiplib.check(L.tn_net_contract(holder.net, hiplib._u64arr([]),
                                           0, None), "leaf net contract")
            local = _Handle(L.tn_net_result_dev(holder.net),
                            int(plan.externals[my_part].size()), holder)

    def send(handle, legs, dims, peer):
        n = handle.elems
        buf = torch.empty((n * 2,), dtype=torch_view, device=device)
        hiplib.check(L.tn_memcpy_dtod(buf.data_ptr(), handle.ptr, n * esize),
                     "tn_memcpy_dtod")
        if backend == "nccl":
            dist_t.send(buf, dst=peer)
        else:  # gloo: host round-trip
            dist_t.send(buf.cpu(), dst=peer)

    def recv(legs, dims, peer):
        n = 1
        for d in dims:
            n *= int(d)
        if backend == "nccl":
            buf = torch.empty((n * 2,), dtype=torch_view, device=device)
            dist_t.recv(buf, src=peer)
        else:
            host = torch.empty((n * 2,), dtype=torch_view)
            dist_t.recv(host, src=peer)
            buf = host.to(device)
        return _Handle(buf.data_ptr(), n, buf)

    def contract_pair(a, a_legs, a_dims, b, b_legs, b_dims):
        net = _PairNet(dev_id, dtype_code)
        ia = L.tn_net_add_leaf_dev(net.net, hiplib._u64arr(a_legs),
                                   hiplib._u64arr(a_dims), len(a_legs),
                                   ctypes.c_void_p(a.ptr))
        ib = L.tn_net_add_leaf_dev(net.net, hiplib._u64arr(b_legs),
                                   hiplib._u64arr(b_dims), len(b_legs),
                                   ctypes.c_void_p(b.ptr))
        assert ia >= 0 and ib >= 0, hiplib.last_error()
        hiplib.check(L.tn_net_contract(net.net, hiplib._u64arr([0, 1]), 1,
                                       None), "pair contract")
        out_elems = 1
        av, bv = set(a_legs), set(b_legs)
        for leg, d in list(zip(a_legs, a_dims)) + list(zip(b_legs, b_dims)):
            if (leg in av) != (leg in bv):
                out_elems *= int(d)
        # tn_net_contract synchronizes before returning, so the input
        # handles (a, b) may be dropped by the caller afterwards
        return _Handle(L.tn_net_result_dev(net.net), out_elems, net)

    return run_fanin(plan, rank, local, send, recv, contract_pair)


def fetch_result(handle, dtype="c128"):
    """Copy a result handle's device buffer to host (flat array)."""
    np_dtype = np.complex128 if dtype == "c128" else np.complex64
    esize = 16 if dtype == "c128" else 8
    out = np.empty(handle.elems, dtype=np_dtype)
    hiplib.check(
        hiplib.lib().tn_memcpy_dtoh(out.ctypes.data_as(ctypes.c_void_p),
                                    ctypes.c_void_p(handle.ptr),
                                    handle.elems * esize),
        "tn_memcpy_dtoh")
    return out
