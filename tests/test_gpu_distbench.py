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
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert lines, (proc.stdout[-2000:], proc.stderr[-2000:])
    d = json.loads(lines[-1])
    assert d["n_gpus"] == 2
    assert d["steps"] == 2
    assert d["config"]["slices"] >= 2
    assert d["value"] > 0
    assert d["roofline"] is not None  # rank-0 dominant-kernel roofline
