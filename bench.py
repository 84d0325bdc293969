#!/usr/bin/env python3
"""Benchmark harness for the MI355X LP hot path.

Driver contract: `python bench.py --gpus N --steps K --warmup W`.
A step = one full deterministic LP refinement (5 sweeps) over the workload
graph from a fixed pseudo-random initial partition (reset between steps),
mirroring the reference's standalone LP benchmark semantics
(apps/benchmarks/shm_label_propagation_benchmark.cc:106-127: LP region only,
allocation/generation excluded).

Workload (BASELINE.json metric: "LP edges processed/sec + final edge-cut,
R-MAT scale-26 k=16"): R-MAT scale-26 (n=2^26, ~1.05G directed arcs,
Graph500 parameters, symmetrized/dedup'd, seed 42), k=16, eps=0.03,
synthetic, unit weights. value = directed arcs scanned per second over the
timed LP regions (whole-job across all ranks).

For N>1 the vertex set of every chunk is sharded across ranks and proposal
lists are all-gathered over RCCL; results are bit-identical to N=1 (see
kaminpar_amd/multi.py).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import numpy as np

HBM_PEAK_GBS = 8000.0  # MI355X spec peak (MI355X_MICROARCH.md)
BYTES_PER_ARC = 8.0    # 4 B adjncy + 4 B labels gather (unweighted model)


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--scale", type=int, default=26, help="R-MAT scale")
    ap.add_argument("--edgefactor", type=int, default=8)
    ap.add_argument("--graph", choices=["rmat", "rgg2d"], default="rmat",
                    help="rgg2d generates the BASELINE config-4 graph "
                         "(n=2^scale, avg degree 16, high locality)")
    ap.add_argument("--k", type=int, default=16)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--workload", choices=["refine", "cluster", "partition"],
                    default="refine",
                    help="cluster mirrors the reference's own LP benchmark "
                         "(shm_label_propagation_benchmark.cc: LP clustering); "
                         "partition runs the progressive-k multilevel "
                         "pipeline (BASELINE config 3; "
                         "kaminpar_amd.partition.partition_deep) -- "
                         "single GPU only")
    ap.add_argument("--order", choices=["natural", "deg-buckets"],
                    default="deg-buckets",
                    help="deg-buckets (default, matching the reference's "
                         "default NodeOrdering::DEGREE_BUCKETS preprocessing) "
                         "reorders vertices before the bench (outside the "
                         "timed region): hub labels become contiguous and "
                         "cache-resident (+13%% at scale 26)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    import kaminpar_amd as ka

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world)
    if args.workload == "partition" and world > 1:
        sys.exit("--workload partition is single-GPU only")

    if world > 1:
        import torch
        import torch.distributed as dist

        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        torch.cuda.set_device(local_rank)
        # force torch's HIP runtime to initialize before the engine's
        # (torch bundles its own libamdhip64; engine-first breaks torch)
        torch.zeros(1, device=f"cuda:{local_rank}")
        # with the C++ RCCL driver (default) the data-path collectives run
        # through the engine's own librccl; torch only coordinates
        # (barriers, the ncclUniqueId broadcast, the elapsed max) -- gloo
        # avoids two RCCL instances contending in one process
        backend = ("gloo" if os.environ.get("KMP_DIST_MODE", "cpp") == "cpp"
                   else "nccl")
        dist.init_process_group(backend)
        device = f"cuda:{local_rank}"
    else:
        device = "cuda:0"

    def generate():
        if args.graph == "rgg2d":
            return ka.Graph.rgg2d(1 << args.scale, avg_deg=16.0, seed=42)
        return ka.Graph.rmat(args.scale, args.edgefactor, seed=42)

    log(f"[bench] generating {args.graph} scale-{args.scale} ...")
    t0 = time.time()
    if world > 1:
        # one generation per node (8 concurrent generator scratches would
        # exhaust host RAM): rank 0 generates + saves, the rest load
        import torch.distributed as dist

        cache = f"/tmp/kmp_{args.graph}{args.scale}_{args.edgefactor}_{args.order}"
        if rank == 0:
            g0 = generate()
            if args.order == "deg-buckets":
                g0, _perm = g0.rearrange_degree_buckets()
            np.save(cache + "_xadj.npy", np.asarray(g0.xadj))
            np.save(cache + "_adjncy.npy", np.asarray(g0.adjncy))
            del g0
        dist.barrier()
        xadj = np.load(cache + "_xadj.npy", mmap_mode="r")
        adjncy = np.load(cache + "_adjncy.npy", mmap_mode="r")
        g = ka.Graph.from_csr(np.asarray(xadj), np.asarray(adjncy))
    else:
        g = generate()
        if args.order == "deg-buckets":
            t1 = time.time()
            g, _perm = g.rearrange_degree_buckets()
            log(f"[bench] deg-bucket rearrangement ({time.time()-t1:.1f}s)")
    log(f"[bench] n={g.n} m={g.m} ({time.time()-t0:.1f}s); uploading ...")

    k = args.k
    part0 = ka.random_partition(g.n, k, seed=5)
    mbw = np.full(k, g.max_block_weight(k, 0.03), dtype=np.int64)
    cut0 = g.edge_cut(part0) if args.workload == "refine" else 0

    eng = ka.LpEngine(g)

    if args.workload == "cluster":
        # max cluster weight per the coarsening formula
        # (max_cluster_weights.h:18-46, EPSILON_BLOCK_WEIGHT, contraction
        # limit 2000, eps=0.03)
        import math
        mcw = int(0.03 * g.total_node_weight / min(max(g.n // 2000, 2), k))

    from kaminpar_amd.multi import (TorchComm, refine_dist,
                                    refine_dist_sharded)

    level_sizes = []
    # KMP_FORCE_DIST=1 exercises the full sharded-commit path (TorchComm +
    # real collectives) at world=1 -- a pre-flight for multi-GPU runs
    force_dist = os.environ.get("KMP_FORCE_DIST") == "1"
    if world > 1 or (force_dist and args.workload == "refine"):
        if world == 1:
            import torch
            import torch.distributed as dist

            torch.zeros(1, device=device)  # torch HIP runtime first
            if not dist.is_initialized():
                os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
                os.environ.setdefault("MASTER_PORT", "29517")
                os.environ.setdefault("RANK", "0")
                os.environ.setdefault("WORLD_SIZE", "1")
                dist.init_process_group("nccl")
        comm = TorchComm(device)
    else:
        comm = None  # fast path: device-resident stepping inside C++
        if args.workload == "refine":
            eng.refine_begin(k, mbw, part0, seed=args.seed)

    def one_step():
        if args.workload == "partition":
            from types import SimpleNamespace

            from kaminpar_amd.partition import partition_deep as ml_partition

            cut, _part, levels, arcs, ns = ml_partition(
                g, k, seed=args.seed, iters=args.iters, engine=eng,
                return_arcs=True)
            level_sizes[:] = levels
            return cut, SimpleNamespace(arcs_scanned=arcs, phase_a_ns=ns,
                                        moves=0)
        if args.workload == "cluster":
            nc, _clus, stats = eng.cluster(mcw, seed=args.seed, iters=args.iters)
            return nc, stats
        if comm is None:
            # timed region: reset (D2D) + sweeps; no host transfers, no cut
            eng.reset()
            eng.run_sweeps(args.iters)
            return None, eng.get_stats()
        # KMP_DIST_MODE: cpp (default; C++ RCCL chunk loop) | sharded
        # (python-orchestrated sharded commit) | replicated (round-1 path)
        mode = os.environ.get("KMP_DIST_MODE", "cpp")
        if mode == "cpp":
            from kaminpar_amd.multi import nccl_cpp_comm, refine_dist_cpp

            nccl_comm = getattr(one_step, "_nccl_comm", None)
            if nccl_comm is None and world > 1:
                nccl_comm = nccl_cpp_comm(rank, world)
                one_step._nccl_comm = nccl_comm
            cut, part, stats = refine_dist_cpp(
                eng, k, mbw, part0, args.seed, args.iters, rank, world,
                nccl_comm)
        else:
            dist_fn = (refine_dist if mode == "replicated"
                       else refine_dist_sharded)
            cut, part, stats = dist_fn(eng, k, mbw, part0, args.seed,
                                       args.iters, comm)
        return cut, stats

    def barrier_sync():
        # single-GPU: eng.refine device-syncs internally; multi-GPU adds the
        # rank barrier + torch-stream sync
        if world > 1:
            import torch
            import torch.distributed as dist

            dist.barrier()
            torch.cuda.synchronize()

    # warmup
    for _ in range(args.warmup):
        one_step()

    barrier_sync()
    t_start = time.time()
    total_arcs = 0
    phase_a_ns = 0
    last_cut = None
    moves = 0
    for _ in range(args.steps):
        cut, stats = one_step()
        total_arcs += stats.arcs_scanned
        phase_a_ns += stats.phase_a_ns
        moves += stats.moves
        last_cut = cut
    barrier_sync()
    t_end = time.time()
    if world == 1 and args.workload == "refine":
        # cut + label download happen once, outside the timed region
        last_cut, _part, _stats = eng.refine_end()

    elapsed = t_end - t_start
    # arcs_scanned counts the WHOLE chunk's processed set and is identical on
    # every rank (labels are replicated), so no reduction is needed; the
    # whole-job rate is arcs / max-over-ranks elapsed.
    if world > 1:
        import torch
        import torch.distributed as dist

        dev = (device if dist.get_backend() == "nccl" else "cpu")
        tmax = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(tmax, op=dist.ReduceOp.MAX)
        elapsed = float(tmax.item())

    value = total_arcs / elapsed

    # roofline of the dominant kernel group (phase A: gain/select+compact):
    # algorithmic bytes 8 B/arc over HIP-event time of those launches. Each
    # rank's phase A scans arcs/world of the work.
    achieved_gbs = (total_arcs / world * BYTES_PER_ARC) / max(phase_a_ns, 1)  # B/ns = GB/s
    # measured HBM traffic per launch of the dominant kernel, from the
    # committed rocprofv3 PMC run (profiles/pmc_traffic.json); null when the
    # workload has no committed measurement
    traffic = None
    try:
        with open(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                               "profiles", "pmc_traffic.json")) as fh:
            tbl = json.load(fh)
        entry = tbl.get(f"rmat{args.scale}_k{args.k}_lp_{args.workload}")
        if entry:
            traffic = entry["fetch_bytes_per_launch"] + entry["write_bytes_per_launch"]
    except OSError:
        pass
    roofline = {
        "bound": "hbm",
        "achieved": round(achieved_gbs, 1),
        "peak": HBM_PEAK_GBS,
        "unit": "GB/s",
        "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
        "traffic": traffic,
    }

    result = None
    if rank == 0:
        cpu_baseline = None
        if not args.no_cpu_baseline and world == 1:
            cpu_baseline = run_cpu_baseline(g, k, mbw, part0, args)

        result = {
            "metric": "LP edges processed/sec",
            "value": round(value, 1),
            "unit": "arcs/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int32",
            "data": "synthetic",
            "config": {
                "workload": (f"{args.graph}{args.scale}_k{args.k}_multilevel"
                             if args.workload == "partition" else
                             f"{args.graph}{args.scale}_k{args.k}_lp_{args.workload}"),
                "n": int(g.n),
                "arcs": int(g.m),
                "k": k,
                "iters": args.iters,
                "edge_cut_before": int(cut0),
                "edge_cut_after": int(last_cut),
                "moves": int(moves // max(args.steps, 1)),
                "parallelism": f"shard{world}" if world > 1 else "single",
                "order": args.order,
                **({"levels": level_sizes} if args.workload == "partition" else {}),
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(result))
    return result


def run_cpu_baseline(g, k, mbw, part0, args):
    """Time the CPU oracle (the restated reference LP semantics, oracle/) on a
    bounded sample of the same workload: ONE sweep (iters=1) over the same
    graph and initial partition (~10-30 s of CPU work at scale 26).

    For the partition workload: time the compiled reference's own serial
    full partitioner (oracle/_ref/libkaminpar_ref_full.so, kind
    "reference") on an R-MAT scale-21 sample (~10 s serial; the full
    serial pipeline on the scale-26 graph extrapolates to ~3-4 min)."""
    if args.workload == "partition":
        return _cpu_baseline_partition(k, args)
    res = _cpu_baseline_impl(g, k, mbw, part0, iters=1, seed=args.seed)
    if res is None:
        return None
    arcs, dt = res
    cores = int(os.environ.get("KMP_ORACLE_THREADS",
                               min(32, os.cpu_count() or 1)))
    return {
        "value": round(arcs / dt, 1),
        "unit": "arcs/s",
        "cores": cores,
        "kind": "port",
        "sample": f"1 LP sweep over the full workload graph ({arcs} arcs, {dt:.1f}s)",
    }


def _cpu_baseline_partition(k, args):
    import ctypes

    here = os.path.dirname(os.path.abspath(__file__))
    path = os.path.join(here, "oracle", "_ref", "libkaminpar_ref_full.so")
    if not os.path.exists(path):
        return None
    import kaminpar_amd as ka

    sample_scale = 21
    gs = ka.Graph.rmat(sample_scale, args.edgefactor, seed=42)
    lib = ctypes.CDLL(path)
    u32p = ctypes.POINTER(ctypes.c_uint32)
    lib.kref_compute_partition.restype = ctypes.c_int64
    xadj = np.ascontiguousarray(gs.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(gs.adjncy, dtype=np.uint32)
    part = np.zeros(gs.n, dtype=np.uint32)
    t0 = time.time()
    cut = lib.kref_compute_partition(
        ctypes.c_uint32(gs.n), ctypes.c_uint64(gs.m),
        xadj.ctypes.data_as(u32p), adjncy.ctypes.data_as(u32p),
        None, None, ctypes.c_uint32(k), ctypes.c_double(0.03),
        ctypes.c_int(args.seed), part.ctypes.data_as(u32p))
    dt = time.time() - t0
    return {
        "value": round(gs.m / dt, 1),
        "unit": "fine arcs partitioned/s",
        "cores": 1,
        "kind": "reference",
        "sample": (f"serial reference KaMinPar::compute_partition on R-MAT "
                   f"scale-{sample_scale} k={k} ({gs.m} arcs, {dt:.1f}s, "
                   f"cut={cut})"),
    }


def _cpu_baseline_impl(g, k, mbw, part0, iters, seed):
    import ctypes

    here = os.path.dirname(os.path.abspath(__file__))
    path = os.path.join(here, "oracle", "liblp_oracle.so")
    if not os.path.exists(path):
        return None
    o = ctypes.CDLL(path)
    o.kmp_oracle_lp_refine.restype = ctypes.c_int64
    part = np.ascontiguousarray(part0, dtype=np.uint32).copy()
    xadj = np.ascontiguousarray(g.xadj, dtype=np.uint32)
    adjncy = np.ascontiguousarray(g.adjncy, dtype=np.uint32)
    stats = np.zeros(3, dtype=np.uint64)
    t0 = time.time()
    o.kmp_oracle_lp_refine(
        ctypes.c_uint32(g.n), ctypes.c_uint64(g.m),
        xadj.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        adjncy.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        None, None, ctypes.c_uint32(k),
        mbw.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        part.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32)),
        ctypes.c_uint64(seed), ctypes.c_int(iters),
        stats.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
    )
    dt = time.time() - t0
    return int(stats[0]), dt


if __name__ == "__main__":
    main()
