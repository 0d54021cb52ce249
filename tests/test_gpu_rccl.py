"""RCCL (`nccl` backend on ROCm) validation on real hardware.

The gloo CPU tests cover the multi-rank orchestration; these cover the
RCCL code path itself (profiles/rccl_probe_r02.md): world=1 communicator
init + device-buffer all_reduce must work, and world=2 with both ranks on
one device must either work (multi-GPU box: ranks land on distinct GPUs)
or fail with RCCL's documented Duplicate-GPU refusal.
"""

import os

import numpy as np
import pytest

torch = pytest.importorskip("torch")
import torch.distributed as dist_t  # noqa: E402
import torch.multiprocessing as mp  # noqa: E402

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        ndev = torch.cuda.device_count()
        torch.cuda.set_device(rank % ndev)
        dist_t.init_process_group("nccl", rank=rank, world_size=world)
        dev = torch.device(f"cuda:{rank % ndev}")
        # c128 viewed as f64 pairs, the bench's wire format
        arr = (np.arange(8) + 1j * np.arange(8)).astype(np.complex128) * (rank + 1)
        t = torch.from_numpy(arr.view(np.float64)).to(dev)
        dist_t.all_reduce(t, op=dist_t.ReduceOp.SUM)
        torch.cuda.synchronize()
        got = t.cpu().numpy().view(np.complex128)
        scale = sum(range(1, world + 1))
        expect = (np.arange(8) + 1j * np.arange(8)).astype(np.complex128) * scale
        ok = np.array_equal(got, expect)
        dist_t.destroy_process_group()
        q.put((rank, "ok" if ok else f"value mismatch: {got}"))
    except Exception as e:
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _run(world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = []
    try:
        for _ in range(world):
            results.append(q.get(timeout=240))
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    return results


def test_rccl_world1_allreduce():
    """The real RCCL branch (device buffers, no host staging) initializes
    and reduces correctly at world=1 — the only size a 1-GPU box allows."""
    results = _run(1, 29541)
    assert results == [(0, "ok")], results


def test_rccl_world2_behavior():
    """On a multi-GPU box two ranks land on distinct devices and the
    collective must be correct; on a 1-GPU box RCCL must refuse with its
    documented Duplicate-GPU error (profiles/rccl_probe_r02.md) — any
    other failure mode is a real bug."""
    results = _run(2, 29542)
    assert len(results) == 2, results
    if torch.cuda.device_count() >= 2:
        assert all(r[1] == "ok" for r in results), results
    else:
        for _, verdict in results:
            assert verdict == "ok" or "Duplicate GPU detected" in verdict, \
                results
