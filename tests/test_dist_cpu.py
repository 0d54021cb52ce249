"""Multi-process distributed contraction logic on CPU: gloo backend,
world_size 2, oracle-backed local contraction. Validates the plan/fan-in
orchestration (scatter semantics, rank mapping, leg bookkeeping, final
forward to rank 0) against the unpartitioned oracle result — the
partitioned == unpartitioned equivalence of integration_tests.rs:26-86
across process boundaries."""

import os

import numpy as np
import pytest

torch = pytest.importorskip("torch")
import torch.distributed as dist_t  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402


def _worker(rank, world, result_queue, nranks_plan):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist_t.init_process_group("gloo", rank=rank, world_size=world)

    from oracle import OTensor, contract_network, contract_tensors
    from oracle.adapters import network_to_otensors
    from tnc_amd.builders import random_circuit
    from tnc_amd.connectivity import ConnectivityLayout
    from tnc_amd.dist import make_plan, run_fanin
    from tnc_amd.tensor import CompositeTensor

    # every rank derives the identical plan deterministically
    tn = random_circuit(13, 8, 0.5, 0.5, 52, ConnectivityLayout.EAGLE)
    plan = make_plan(tn, nranks_plan, trials=4)

    my_part = None
    for p, r in plan.part_rank.items():
        if r == rank:
            my_part = p

    local = None
    if my_part is not None:
        sub = plan.partitioned.tensors[my_part]
        if isinstance(sub, CompositeTensor):
            inner = plan.path.nested.get(my_part)
            otensors = network_to_otensors(sub)
            if inner is not None and inner.toplevel:
                local = contract_network(otensors, inner)
            else:
                assert len(otensors) == 1
                local = otensors[0]
        else:
            from oracle.adapters import leaf_to_otensor

            local = leaf_to_otensor(sub)

    def send(handle, legs, dims, peer):
        arr = np.ascontiguousarray(handle.data, dtype=np.complex128)
        t = torch.from_numpy(arr.view(np.float64).reshape(-1))
        dist_t.send(t, dst=peer)

    def recv(legs, dims, peer):
        n = int(np.prod(dims)) if dims else 1
        t = torch.empty(n * 2, dtype=torch.float64)
        dist_t.recv(t, src=peer)
        data = t.numpy().view(np.complex128).reshape([int(d) for d in dims])
        return OTensor(list(legs), data)

    def contract_pair(a, a_legs, a_dims, b, b_legs, b_dims):
        assert list(a.legs) == list(a_legs)
        assert list(b.legs) == list(b_legs)
        return contract_tensors(a, b)

    final = run_fanin(plan, rank, local, send, recv, contract_pair)
    if rank == 0:
        ref = contract_network(
            network_to_otensors(tn),
            __import__("tnc_amd").RandomGreedy(4).find_path(tn).replace_path(),
        )
        ok = np.allclose(final.data, ref.data, rtol=1e-10, atol=1e-12)
        result_queue.put(("ok" if ok else
                          f"mismatch: {final.data} vs {ref.data}"))
    dist_t.destroy_process_group()


def _run(world, nranks_plan):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, q, nranks_plan))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    try:
        verdict = q.get(timeout=180)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    assert verdict == "ok", verdict


def test_dist_fanin_two_ranks():
    _run(2, 2)


def test_dist_fanin_four_parts_two_unused_ranks():
    # plan for 4 ranks executed on 4 processes; exercises rank mapping with
    # the final partition pinned to rank 0 (communication.rs:89-115)
    _run(4, 4)


def test_make_plan_rqc36_eight_way():
    """The benchmark's 8-way plan builds deterministically and respects the
    rank mapping contract (final partition -> rank 0)."""
    from tnc_amd.dist import make_plan
    from tnc_amd.fixtures import load_fixture

    tn, _, _ = load_fixture("rqc36")
    plan = make_plan(tn, 8, trials=4, size_cap=4.0e9)
    assert plan.nparts == 8
    assert plan.used_ranks <= 8
    assert sorted(plan.part_rank.values()) == list(range(plan.used_ranks))
    final_part = plan.path.toplevel[-1][0]
    assert plan.part_rank[final_part] == 0
    assert len(plan.path.toplevel) == plan.nparts - 1
    assert plan.total_flops() > 0
    # deterministic across "ranks"
    plan2 = make_plan(tn, 8, trials=4, size_cap=4.0e9)
    assert plan2.part_rank == plan.part_rank
    assert plan2.path.toplevel == plan.path.toplevel


def test_tree_plan_fanin_two_ranks_gloo():
    """Tree-split plan executed across 2 gloo processes == direct result."""
    _run_tree(2)


def _tree_worker(rank, world, result_queue):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29519"
    dist_t.init_process_group("gloo", rank=rank, world_size=world)

    import numpy as np

    from oracle import OTensor, contract_network, contract_tensors
    from oracle.adapters import network_to_otensors
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.dist import make_tree_plan, run_fanin
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.tensor import CompositeTensor

    tn, rp, _ = load_fixture("rqc24")
    plan = make_tree_plan(tn, rp, world)

    my_part = None
    for p, r in plan.part_rank.items():
        if r == rank:
            my_part = p
    local = None
    if my_part is not None:
        sub = plan.partitioned.tensors[my_part]
        inner = plan.path.nested.get(my_part)
        otensors = network_to_otensors(sub)
        if inner is not None and inner.toplevel:
            local = contract_network(otensors, inner)
        else:
            assert len(otensors) == 1
            local = otensors[0]

    def send(handle, legs, dims, peer):
        arr = np.ascontiguousarray(handle.data, dtype=np.complex128)
        t = torch.from_numpy(arr.view(np.float64).reshape(-1))
        dist_t.send(t, dst=peer)

    def recv(legs, dims, peer):
        n = int(np.prod(dims)) if dims else 1
        t = torch.empty(n * 2, dtype=torch.float64)
        dist_t.recv(t, src=peer)
        data = t.numpy().view(np.complex128).reshape([int(d) for d in dims])
        return OTensor(list(legs), data)

    def contract_pair(a, a_legs, a_dims, b, b_legs, b_dims):
        assert list(a.legs) == list(a_legs)
        return contract_tensors(a, b)

    final = run_fanin(plan, rank, local, send, recv, contract_pair)
    if rank == 0:
        ref = contract_network(network_to_otensors(tn),
                               ContractionPath.simple(rp))
        ok = np.allclose(final.data, ref.data, rtol=1e-12)
        result_queue.put("ok" if ok else "mismatch")
    dist_t.destroy_process_group()


def _run_tree(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tree_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    try:
        verdict = q.get(timeout=180)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    assert verdict == "ok", verdict


def test_tree_plan_fanin_four_ranks_gloo():
    _run_tree(4)


def test_tree_plan_rqc36_shapes():
    """Tree plans preserve total flops and bound exchanged tensors by the
    frozen path's own intermediates at every N."""
    from tnc_amd.dist import make_tree_plan
    from tnc_amd.fixtures import load_fixture

    for nr in (2, 4, 8):
        tn, rp, meta = load_fixture("rqc36")
        plan = make_tree_plan(tn, rp, nr)
        assert abs(plan.total_flops() - meta["metric_flops"]) < 1e6
        for ext in plan.externals:
            assert ext.size() <= meta["peak_size_elems"]
        assert len(plan.path.toplevel) == plan.nparts - 1


def _slice_worker(rank, world, result_queue):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29519"
    dist_t.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import math

        from oracle import contract_network
        from oracle.adapters import network_to_otensors
        from tnc_amd.contraction_path import ContractionPath
        from tnc_amd.fixtures import load_fixture
        from tnc_amd.slicing import (find_slice_edges, iter_assignments,
                                     slice_network)

        tn, rp, meta = load_fixture("rqc24")
        replace = ContractionPath.simple(rp)
        ebits = max(1, int(math.ceil(math.log2(world))))
        edges, _ = find_slice_edges(tn, rp, 0, max_edges=ebits)
        assignments = list(iter_assignments(tn, edges))
        assert len(edges) == ebits, (edges, ebits)
        # the bench's rank->slice mapping: rank r takes assignments[r::world]
        local = None
        for a in assignments[rank::world]:
            stn = slice_network(tn, a)
            part = contract_network(network_to_otensors(stn), replace)
            arr = np.atleast_1d(np.asarray(part.data, dtype=np.complex128))
            local = arr.copy() if local is None else local + arr
        if local is None:
            local = np.zeros(1, dtype=np.complex128)
        buf = torch.from_numpy(local.view(np.float64))
        dist_t.all_reduce(buf, op=dist_t.ReduceOp.SUM)
        total = buf.numpy().view(np.complex128)
        if rank == 0:
            direct = contract_network(network_to_otensors(tn), replace)
            np.testing.assert_allclose(
                total, np.atleast_1d(direct.data), rtol=1e-10, atol=1e-14)
            result_queue.put("ok")
    except Exception as e:  # pragma: no cover
        if rank == 0:
            result_queue.put(f"FAIL: {e!r}")
        raise
    finally:
        dist_t.destroy_process_group()


def _run_sliced(world):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_slice_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    try:
        verdict = q.get(timeout=180)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    assert verdict == "ok", verdict


def test_sliced_allreduce_two_ranks_gloo():
    """The multi-GPU bench's slicing scheme on CPU: ranks contract their
    slice assignments, one all_reduce sums them; equals the direct
    contraction."""
    _run_sliced(2)


def test_sliced_allreduce_four_ranks_gloo():
    """Four ranks, two sliced edges (the N=4 shape of the scale run)."""
    _run_sliced(4)
