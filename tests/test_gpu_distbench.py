"""Rehearse the driver's multi-GPU bench launch on one device.

The scale run executes `python -m torch.distributed.run --nnodes=1
--nproc-per-node N ... bench.py --gpus N ...`. This test runs that exact
command with N=2 on the single available GPU (both ranks map to device 0
via the modulo device assignment) using the gloo backend for the
collective (RCCL refuses two ranks on one device,
profiles/rccl_probe_r02.md) — real ContractionEngines, real slicing,
real all_reduce plumbing, real rendezvous. Asserts rank 0 emits the JSON
line with the distributed metadata.
"""

import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_distributed_bench_two_ranks_one_gpu():
    env = dict(os.environ)
    env["TN_BENCH_BACKEND"] = "gloo"
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", "29551", os.path.join(ROOT, "bench.py"),
        "--gpus", "2", "--steps", "2", "--warmup", "1",
        "--fixture", "rqc24", "--no-secondary", "--no-cpu-baseline",
    ]
    proc = subprocess.run(cmd, cwd=ROOT, env=env, capture_output=True,
                          text=True, timeout=420)
    assert proc.returncode == 0, (proc.stdout[-2000:], proc.stderr[-2000:])
    lines = [json.loads(l) for l in proc.stdout.splitlines()
             if l.startswith("{")]
    assert len(lines) == 2, (proc.stdout[-2000:], proc.stderr[-2000:])
    fanin, headline = lines
    # secondary mechanism line: partition fan-in (communication.rs shape)
    assert fanin.get("secondary") is True
    assert fanin["config"]["mechanism"] == "partition-fanin"
    assert fanin["n_gpus"] == 2 and fanin["value"] > 0
    # headline: edge slicing with rank-0 roofline
    d = headline
    assert d["n_gpus"] == 2
    assert d["steps"] == 2
    assert d["config"]["slices"] >= 2
    assert d["value"] > 0
    assert d["roofline"] is not None  # rank-0 dominant-kernel roofline


def _fanin_worker(rank, world, q):
    import numpy as np
    import torch
    import torch.distributed as dist_t

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29553"
    torch.cuda.set_device(0)
    dist_t.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from oracle import contract_network
        from oracle.adapters import network_to_otensors
        from tnc_amd.dist import make_tree_plan
        from tnc_amd.dist_gpu import fetch_result, run_fanin_gpu
        from tnc_amd.fixtures import load_fixture

        tn, rp, meta = load_fixture("rqc24")
        plan = make_tree_plan(tn, rp, world)
        handle = run_fanin_gpu(plan, rank, world, dist_t, torch,
                               torch.device("cuda:0"), dev_id=0,
                               dtype="c128", backend="gloo")
        if rank == 0:
            got = fetch_result(handle, "c128")
            ref = contract_network(network_to_otensors(tn), rp)
            ok = np.allclose(got.reshape(-1),
                             np.atleast_1d(ref.data).reshape(-1),
                             rtol=1e-10, atol=1e-14)
            q.put("ok" if ok else f"mismatch {got} vs {ref.data}")
    except Exception as e:  # pragma: no cover
        if rank == 0:
            q.put(f"FAIL: {e!r}")
        raise
    finally:
        dist_t.destroy_process_group()


def test_gpu_fanin_two_ranks_vs_oracle():
    """The GPU fan-in backend (real engines, device-buffer exchange,
    tn_net pair merges) reproduces the oracle across 2 process ranks on
    one device (gloo wire; the nccl branch differs only in skipping the
    host round-trip)."""
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_fanin_worker, args=(r, 2, q))
             for r in range(2)]
    for p in procs:
        p.start()
    try:
        verdict = q.get(timeout=240)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    assert verdict == "ok", verdict
