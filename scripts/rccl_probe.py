#!/usr/bin/env python3
"""Probe: can RCCL form a 2-rank communicator with both ranks on ONE GPU?

VERDICT r01 item 3: the gloo CPU tests never execute the RCCL code path;
before an 8-GPU node shows up, validate the `nccl`-backend branch on a
single-GPU lease. Runs three experiments and prints one [RCCL-PROBE] line
each:
  1. world=2 on one device: all_reduce of a c128-viewed tensor
  2. world=2 on one device: send/recv pair (the fan-in wire pattern)
  3. world=1: degenerate RCCL all_reduce (communicator init + collective)
"""

import os
import sys
import traceback

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        torch.cuda.set_device(0)
        dist.init_process_group("nccl", rank=rank, world_size=world)
        dev = torch.device("cuda:0")
        # 1. all_reduce (the slicing mechanism's collective)
        t = torch.full((8, 2), float(rank + 1), dtype=torch.float64,
                       device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        torch.cuda.synchronize()
        expect = sum(range(1, world + 1))
        ok_ar = bool((t == expect).all().item())
        # 2. send/recv (the fan-in wire pattern), world 2 only
        ok_sr = None
        if world == 2:
            if rank == 1:
                payload = torch.arange(16, dtype=torch.float64,
                                       device=dev) * 0.5
                dist.send(payload, dst=0)
            else:
                buf = torch.empty(16, dtype=torch.float64, device=dev)
                dist.recv(buf, src=1)
                torch.cuda.synchronize()
                ok_sr = bool(torch.allclose(
                    buf, torch.arange(16, dtype=torch.float64,
                                      device=dev) * 0.5))
        dist.destroy_process_group()
        q.put((rank, "ok", ok_ar, ok_sr))
    except Exception as e:
        q.put((rank, f"FAIL: {type(e).__name__}: {e}", None, None))


def run(world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = []
    try:
        for _ in range(world):
            results.append(q.get(timeout=180))
    except Exception:
        traceback.print_exc()
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    return results


def main():
    print(f"[RCCL-PROBE] torch {torch.__version__}, "
          f"devices={torch.cuda.device_count()}, "
          f"nccl_version={torch.cuda.nccl.version()}")
    r2 = run(2, 29531)
    print(f"[RCCL-PROBE] world=2 one-GPU: {sorted(r2)}")
    r1 = run(1, 29532)
    print(f"[RCCL-PROBE] world=1: {sorted(r1)}")
    ok2 = all(x[1] == "ok" and x[2] for x in r2) and len(r2) == 2
    ok1 = all(x[1] == "ok" and x[2] for x in r1) and len(r1) == 1
    print(f"[RCCL-PROBE] verdict: two_ranks_one_gpu={'OK' if ok2 else 'REFUSED/FAILED'} "
          f"single_rank={'OK' if ok1 else 'FAILED'}")
    sys.exit(0 if (ok2 or ok1) else 1)


if __name__ == "__main__":
    main()
