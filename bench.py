#!/usr/bin/env python3
"""Benchmark: pairwise-contraction GFLOP/s (c128) on the frozen 36-qubit RQC
amplitude network (BASELINE.json configs[2]; --gpus N>1 = configs[3],
one partition per GPU over RCCL/xGMI).

Protocol mirrors the reference benchmark (benchmark/src/main.rs:355-405):
the contraction plan (path / partitioning) is precomputed (frozen in the
fixture / derived deterministically) and excluded from timing; the clock
runs from the pre-step barrier to the final tensor on GPU 0. A "step" is one
full contraction of the fixture network. value = executed metric flops
(sum over path steps of (8*s-2)*o, contraction_cost.rs:26-32) / wall time,
aggregated over all ranks.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--fixture rqc36]
Multi-GPU: launched via torch.distributed.run, one rank per GPU (RCCL).
"""

import argparse
import json
import os
import sys
import time

# cap BLAS threads before numpy loads (cpu_baseline; see _blas_threads)
os.environ.setdefault("OPENBLAS_NUM_THREADS", "64")
os.environ.setdefault("OMP_NUM_THREADS", "64")

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

# MI355X peaks (MI355X_MICROARCH.md; f64 matrix peak cross-checked by the
# committed muBench under profiles/ — see DESIGN.md "Measurement")
F64_MFMA_PEAK = 78.6e12   # real flops/s, v_mfma_f64_16x16x4_f64 dense
F32_MFMA_PEAK = 157.3e12  # real flops/s, v_mfma_f32_16x16x4_f32 (c64 path)
HBM_PEAK = 8.0e12         # bytes/s (spec; ~6.3 TB/s achievable)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--fixture", default="rqc36")
    p.add_argument("--no-cpu-baseline", action="store_true")
    p.add_argument("--no-secondary", action="store_true",
                   help="skip the rqc24/syc49 secondary evidence lines")
    p.add_argument("--trials", type=int, default=16,
                   help="random-greedy trials for the multi-GPU plan")
    return p.parse_args()


def step_bytes(info, esize=16):
    """Algorithmic bytes of one einsum step: read A and B once, write out
    once (esize bytes per element)."""
    return float(esize) * (info.m * info.k + info.k * info.n + info.m * info.n)


def roofline_from_profile(infos, step_ms, gemm_ms, kinds, dtype="c128",
                          fixture="rqc36"):
    """Dominant kernel + its roofline leg from live HIP-event timings.
    `traffic` comes from the committed rocprofv3 --pmc calibration when one
    exists for THIS fixture (profiles/pmc_calibration.json), else null."""
    dom = max(range(len(step_ms)), key=lambda s: step_ms[s])
    info = infos[dom]
    traffic = None
    pmc_path = os.path.join(ROOT, "profiles", "pmc_calibration.json")
    if os.path.exists(pmc_path):
        try:
            with open(pmc_path) as f:
                pmc = json.load(f)
            entry = pmc.get("fixtures", {}).get(fixture)
            if entry:
                key = f"{int(info.m)}x{int(info.n)}x{int(info.k)}"
                traffic = entry.get("by_mnk", {}).get(
                    key, entry.get("dominant_kernel_traffic_bytes"))
        except Exception:
            traffic = None
    mfma_peak = F64_MFMA_PEAK if dtype == "c128" else F32_MFMA_PEAK
    esize = 16 if dtype == "c128" else 8
    if kinds[dom] >= 2 and gemm_ms[dom] > 0:
        dur_s = gemm_ms[dom] / 1e3
        achieved = info.flops / dur_s
        return {
            "bound": "mfma",
            "achieved": achieved,
            "peak": mfma_peak,
            "unit": "FLOP/s",
            "frac": achieved / mfma_peak,
            "traffic": traffic,
            "kernel": ("k_zgemm_c128_glds_pure" if dtype == "c128"
                       else "k_zgemm_c64_glds_pure"),
            "launch_ms": gemm_ms[dom],
            "mnk": [info.m, info.n, info.k],
        }
    dur_s = step_ms[dom] / 1e3
    achieved = step_bytes(info, esize) / dur_s
    return {
        "bound": "hbm",
        "achieved": achieved,
        "peak": HBM_PEAK,
        "unit": "B/s",
        "frac": achieved / HBM_PEAK,
        "traffic": traffic,
        "kernel": "k_einsum_smallk",
        "launch_ms": step_ms[dom],
        "mnk": [info.m, info.n, info.k],
    }


def _blas_threads():
    try:
        cores = len(os.sched_getaffinity(0))
    except AttributeError:
        cores = os.cpu_count() or 1
    # OpenBLAS misbehaves (and slows down) beyond ~64 threads; the thread
    # count actually used is reported in the JSON ("cores")
    return min(64, cores)


def _mem_available_bytes():
    try:
        with open("/proc/meminfo") as f:
            for line in f:
                if line.startswith("MemAvailable"):
                    return int(line.split()[1]) * 1024
    except OSError:
        pass
    return None


def oracle_peak_bytes(tn, replace_toplevel, esize=16):
    """Peak live bytes of the oracle's replace-left walk (all live slots +
    the step's output), from metadata only."""
    from tnc_amd.tensor import LeafTensor

    views = [LeafTensor(t.legs, t.bond_dims) for t in tn.tensors]
    live = {i: v.size() for i, v in enumerate(views)}
    peak = 0.0
    for i, j in replace_toplevel:
        out = views[i] ^ views[j]
        peak = max(peak, sum(live.values()) + out.size())
        del live[j]
        live[i] = out.size()
        views[i] = out
        views[j] = None
    return peak * esize


def cpu_baseline_full(fixture, tn, replace_toplevel, metric_flops):
    """The reference benchmark's own protocol (benchmark/src/main.rs:355-363):
    time the oracle contracting the FULL network along the same frozen path
    (path excluded from timing), value = the metric numerator / wall.
    Returns None when the walk won't fit host RAM (caller falls back to the
    sampled estimate). kind="port" (the oracle is our restatement, not the
    reference binary)."""
    import oracle
    from oracle.adapters import network_to_otensors

    # np.einsum's TTGT materializes transposed operand copies on the heavy
    # steps — budget 1.7x the walk's live peak before trusting RAM
    avail = _mem_available_bytes()
    need = oracle_peak_bytes(tn, replace_toplevel) * 1.7
    if avail is not None and need > avail:
        return None
    ots = network_to_otensors(tn)
    t0 = time.perf_counter()
    oracle.contract_network(ots, replace_toplevel)
    dt = time.perf_counter() - t0
    return {
        "value": metric_flops / dt / 1e9,
        "unit": "GFLOP/s",
        "cores": _blas_threads(),
        "kind": "port",
        "sample": f"full {fixture} network, all {len(replace_toplevel)} "
                  f"frozen-path steps contracted by the oracle (numpy einsum "
                  f"-> host BLAS zgemm), {dt:.1f}s wall",
    }


def cpu_baseline(infos, fixture="rqc36", budget_s=20.0, cap_elems=2 ** 29,
                 dtype="c128"):
    """Oracle (numpy einsum -> BLAS zgemm) timed on the host cores over a
    bounded sample: the largest path steps whose operands fit `cap_elems`,
    random-valued inputs of the same shapes (einsum time is value-
    independent), until ~budget_s of wall. kind="port"."""
    import numpy as np

    import oracle

    all_flops = sum(i.flops for i in infos)
    order = sorted(range(len(infos)), key=lambda s: -infos[s].flops)
    rng = np.random.default_rng(0)
    total_flops = 0.0
    total_time = 0.0
    used = 0
    shrunk = 0
    for s in order:
        info = infos[s]
        # steps too large for host RAM are sampled at reduced M (same K, N:
        # the per-flop gemm rate is M-extensive, so the measured rate stands
        # in for the full step; flops credited are the SAMPLED ones)
        m, n, k = int(info.m), int(info.n), int(info.k)
        was_shrunk = False
        while m > 1 and (m * k + k * n + m * n) > cap_elems:
            m //= 2
            was_shrunk = True
        if (m * k + k * n + m * n) > cap_elems:
            continue
        shrunk += was_shrunk
        npdtype = np.complex128 if dtype == "c128" else np.complex64
        a = (rng.standard_normal((m, k)) + 1j * rng.standard_normal((m, k))
             ).astype(npdtype)
        b = (rng.standard_normal((k, n)) + 1j * rng.standard_normal((k, n))
             ).astype(npdtype)
        t0 = time.perf_counter()
        oracle.contract_ndarrays([0, 2], [0, 1], a, [1, 2], b)
        dt = time.perf_counter() - t0
        total_time += dt
        # flops of the sampled (possibly reduced-M) gemm
        total_flops += (8.0 * info.k - 2.0) * m * n
        used += 1
        if total_time >= budget_s:
            break
    if total_time == 0.0:
        return None
    return {
        "value": total_flops / total_time / 1e9,
        "unit": "GFLOP/s",
        "cores": _blas_threads(),
        "kind": "port",
        "sample": f"{used} largest {fixture} path steps ({shrunk} sampled "
                  f"at reduced M to fit {cap_elems} elems), random-valued "
                  f"same-(N,K) {'zgemm' if dtype == 'c128' else 'cgemm'}, "
                  f"{total_time:.1f}s, {total_flops:.3e} sampled flops",
    }


def emit(result):
    print(json.dumps(result), flush=True)


# workload descriptions (BASELINE.json configs; path provenance is in the
# fixture meta, surfaced under config.path_provenance)
WORKLOADS = {
    "rqc36": "rqc36: 36q depth-14 RQC single-amplitude network, frozen "
             "PartitionSearch path (BASELINE config 3 headline)",
    "rqc24": "rqc24: 24q depth-10 RQC single-amplitude network, frozen "
             "PartitionSearch path (BASELINE config 2; launch/replay bound)",
    "syc49": "syc49: 49q Sycamore-style RQC single amplitude, c64, frozen "
             "PartitionSearch path (BASELINE config 5)",
}


def run_single(args, fixture=None, steps=None, warmup=None, secondary=False):
    from tnc_amd.contraction_path import ContractionPath
    from tnc_amd.executor import ContractionEngine
    from tnc_amd.fixtures import load_fixture

    fixture = fixture or args.fixture
    steps = args.steps if steps is None else steps
    warmup = args.warmup if warmup is None else warmup
    tn, replace_toplevel, meta = load_fixture(fixture)
    dtype = meta.get("dtype", "c128")
    replace = ContractionPath.simple(replace_toplevel)
    eng = ContractionEngine(tn, replace, device=0, dtype=dtype)
    flops_per_contraction = eng.total_flops

    # one profiled pass (doubles as extra warmup)
    _, step_ms, gemm_ms, kinds = eng.contract_profiled()
    roofline = roofline_from_profile(eng.infos, step_ms, gemm_ms, kinds, dtype,
                                     fixture)

    for _ in range(warmup):
        eng.contract()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng.contract()
    wall = time.perf_counter() - t0

    value = flops_per_contraction * steps / wall / 1e9
    infos = eng.infos
    eng.close()  # release the device arena before the host-BLAS baseline
    cb = None
    if not args.no_cpu_baseline:
        if dtype == "c128":
            # reference protocol: full-network oracle contraction (falls
            # back to the sampled estimate when host RAM is short)
            cb = cpu_baseline_full(fixture, tn, replace_toplevel,
                                   flops_per_contraction)
        if cb is None:
            cb = cpu_baseline(infos, fixture, dtype=dtype)
    config = {
        "workload": WORKLOADS.get(fixture, fixture),
        "tensors": len(tn.tensors),
        "path_steps": len(replace_toplevel),
        "metric_flops_per_contraction": flops_per_contraction,
    }
    if meta.get("path_finder"):
        config["path_provenance"] = meta["path_finder"]
    line = {
        "metric": f"pairwise-contraction GFLOP/s ({dtype})",
        "value": value,
        "unit": "GFLOP/s",
        "n_gpus": 1,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": wall / steps * 1e3,
        "higher_is_better": True,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": dtype,
        "data": "synthetic",
        "config": config,
        "roofline": roofline,
        "cpu_baseline": cb,
    }
    if secondary:
        line["secondary"] = True
    emit(line)


def _fanin_secondary(args, tn, frozen_path, meta, rank, world, dev_id,
                     device, backend, dist_t, torch, useful_flops, dtype):
    """Partition fan-in mechanism (tree-cut plan, one partition per rank,
    P2P exchange of open-leg intermediates, final on rank 0). The timed
    region includes per-iteration engine setup (leaf scatter) like the
    reference's in-run scatter; tree-cut leaves the heavy top-of-tree
    merges sequential, so this line documents WHY slicing is the headline
    mechanism."""
    from tnc_amd.dist import make_tree_plan
    from tnc_amd.dist_gpu import run_fanin_gpu

    plan = make_tree_plan(tn, frozen_path, world)
    steps = max(1, min(args.steps, 3))
    warmup = 1
    times = []
    for it in range(warmup + steps):
        dist_t.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        handle = run_fanin_gpu(plan, rank, world, dist_t, torch, device,
                               dev_id=dev_id, dtype=dtype, backend=backend)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        del handle
        dt_t = torch.tensor([dt], dtype=torch.float64,
                            device=device if backend == "nccl" else "cpu")
        dist_t.all_reduce(dt_t, op=dist_t.ReduceOp.MAX)
        if it >= warmup:
            times.append(dt_t.item())
    if rank == 0:
        wall = sum(times)
        emit({
            "metric": f"pairwise-contraction GFLOP/s ({dtype})",
            "value": useful_flops * steps / wall / 1e9,
            "unit": "GFLOP/s",
            "n_gpus": world,
            "steps": steps,
            "warmup": warmup,
            "ms_per_step": wall / steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic",
            "config": {
                "workload": f"{args.fixture}: partition fan-in mechanism "
                            "(communication.rs:199-249 semantics; tree-cut "
                            f"{plan.nparts}-way plan, per-iteration scatter "
                            "included in the timed region)",
                "mechanism": "partition-fanin",
                "partitions": plan.nparts,
                "metric_flops_per_contraction": useful_flops,
            },
            "roofline": None,
            "cpu_baseline": None,
            "secondary": True,
        })


def run_distributed(args):
    """N ranks, one GPU each (torch.distributed / RCCL). Parallelism is
    EDGE SLICING (tnc_amd/slicing.py): ceil(log2(N)) shared edges of the
    frozen path's near-peak intermediates are fixed per slice, every rank
    contracts its own slice(s) of the full network independently (same
    replace-left path, hipGraph-replayed), and one RCCL all_reduce sums
    the slice results. A tree-cut of the frozen contraction path (see
    tnc_amd/dist.py, kept as the reference's MPI fan-in mirror) measures
    ~1-2x at N=2-8 because the heavy top-of-tree merges are sequential;
    sliced contraction has 93.7%% flops-ideal efficiency at N=8 on rqc36
    (6.7%% slicing overhead, counted against `value` — the numerator is
    the UNSLICED path's metric flops, so `value` reflects real speedup).
    """
    import math as _math

    import torch
    import torch.distributed as dist_t

    from tnc_amd import hiplib
    from tnc_amd.contraction_path import ContractionPath, flatten_network
    from tnc_amd.executor import ContractionEngine, plan_steps
    from tnc_amd.fixtures import load_fixture
    from tnc_amd.slicing import (find_slice_edges, iter_assignments,
                                 slice_network)

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    # modulo lets a 1-GPU box smoke-test the N-rank path (the real scale
    # run has one GPU per local rank, where this is the identity)
    dev_id = local_rank % max(1, torch.cuda.device_count())
    torch.cuda.set_device(dev_id)
    backend = os.environ.get("TN_BENCH_BACKEND", "nccl")
    dist_t.init_process_group(backend)
    device = torch.device(f"cuda:{dev_id}")

    tn, frozen_path, meta = load_fixture(args.fixture)
    dtype = meta.get("dtype", "c128")
    torch_view = torch.float64 if dtype == "c128" else torch.float32
    replace = ContractionPath.simple(frozen_path)

    # useful (unsliced) flops: the metric numerator, constant across N
    leaves, steps, _ = flatten_network(tn, replace)
    plan_infos = plan_steps(leaves, steps)
    useful_flops = sum(i.flops for i in plan_infos)

    # SECONDARY mechanism line: partition fan-in (the reference's MPI
    # shape, communication.rs:199-249) measured beside the slicing
    # headline so an 8-GPU run records both. Fail-safe: any error here
    # only drops this line, never the headline.
    if not os.environ.get("TN_NO_FANIN"):
        try:
            _fanin_secondary(args, tn, frozen_path, meta, rank, world,
                             dev_id, device, backend, dist_t, torch,
                             useful_flops, dtype)
        except Exception as e:  # noqa: BLE001
            print(f"[bench] fan-in secondary failed: {e!r}",
                  file=sys.stderr, flush=True)
    # final-tensor element count from the PLAN metadata (identical on every
    # rank, sliced or not — sliced edges are internal, the final view is the
    # network's): ranks whose assignment share is empty still allocate the
    # same all_reduce buffer shape.
    if plan_infos:
        elems = 1
        for d in plan_infos[-1].out_dims:
            elems *= int(d)
    else:
        elems = int(leaves[0].size()) if leaves else 1

    ebits = max(1, int(_math.ceil(_math.log2(world))))
    edges, _peak = find_slice_edges(tn, frozen_path, 0, max_edges=ebits)
    assignments = list(iter_assignments(tn, edges))
    mine = assignments[rank::world]
    engines = []
    for a in mine:
        stn = slice_network(tn, a)
        engines.append(
            ContractionEngine(stn, replace, device=dev_id, dtype=dtype))
    executed_flops = (sum(e.total_flops for e in engines[:1]) *
                      len(assignments)) if engines else 0.0

    # rank 0's dominant-kernel roofline (one profiled pass, untimed; also
    # extra warmup) so N>1 lines carry the roofline axis
    roofline = None
    if rank == 0 and engines:
        _, p_step_ms, p_gemm_ms, p_kinds = engines[0].contract_profiled()
        roofline = roofline_from_profile(
            engines[0].infos, p_step_ms, p_gemm_ms, p_kinds, dtype,
            args.fixture)
    # one untimed pass per engine to arm graph capture
    for eng in engines:
        eng.contract()
    L = hiplib.lib()

    local = torch.zeros((elems, 2), dtype=torch_view, device=device)
    tmp = torch.empty((elems, 2), dtype=torch_view, device=device)
    times = []
    for it in range(args.warmup + args.steps):
        dist_t.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        local.zero_()
        for eng in engines:
            eng.contract()
            hiplib.check(
                L.tn_memcpy_dtod(tmp.data_ptr(), eng.result_dev(),
                                 elems * (16 if dtype == "c128" else 8)),
                "tn_memcpy_dtod")
            local += tmp
        if backend == "nccl":
            dist_t.all_reduce(local, op=dist_t.ReduceOp.SUM)
        else:  # gloo (CPU-only collectives): testing path
            host = local.cpu()
            dist_t.all_reduce(host, op=dist_t.ReduceOp.SUM)
            local.copy_(host)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        dt_t = torch.tensor([dt], dtype=torch.float64,
                            device=device if backend == "nccl" else "cpu")
        dist_t.all_reduce(dt_t, op=dist_t.ReduceOp.MAX)
        if it >= args.warmup:
            times.append(dt_t.item())

    if rank == 0:
        wall = sum(times)
        value = useful_flops * args.steps / wall / 1e9
        emit({
            "metric": f"pairwise-contraction GFLOP/s ({dtype})",
            "value": value,
            "unit": "GFLOP/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": wall / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": dtype,
            "data": "synthetic",
            "config": {
                "workload": f"{args.fixture}: 36q depth-14 RQC amplitude, "
                            f"{len(assignments)}-way edge slicing, RCCL "
                            "all_reduce sum"
                            if args.fixture == "rqc36" else
                            f"{args.fixture}, {len(assignments)}-way edge "
                            "slicing",
                "tensors": len(tn.tensors),
                "sliced_edges": len(edges),
                "slices": len(assignments),
                "metric_flops_per_contraction": useful_flops,
                "executed_flops_per_contraction": executed_flops,
                "slice_overhead": (executed_flops / useful_flops
                                   if useful_flops else None),
            },
            "roofline": roofline,
            # cpu_baseline is measured on rank 0 at N=1 only (see run_single)
            "cpu_baseline": None,
        })
    for eng in engines:
        eng.close()
    dist_t.destroy_process_group()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        run_distributed(args)
        return
    # Secondary lines for the other BASELINE configs (rqc24 = config 2,
    # syc49 = config 5/c64), emitted BEFORE the headline so the headline is
    # the last JSON line on stdout. Few steps each — they are evidence
    # lines, not the headline measurement.
    if args.fixture == "rqc36" and not args.no_secondary:
        run_single(args, fixture="rqc24", steps=min(args.steps, 10),
                   warmup=min(args.warmup, 2), secondary=True)
        run_single(args, fixture="syc49", steps=min(args.steps, 5),
                   warmup=min(args.warmup, 2), secondary=True)
    run_single(args)


if __name__ == "__main__":
    main()
