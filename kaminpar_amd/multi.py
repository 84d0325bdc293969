"""Multi-GPU sharded LP refinement: one process per GPU over RCCL.

Design (mirrors the reference's distributed LP round structure,
kaminpar-dist/coarsening/clustering/lp/global_lp_clusterer.cc:160-191 and
refinement/lp/lp_refiner.cc:296-333, with labels REPLICATED instead of
owner-sharded -- a scale-26 label array is 256 MB, trivially replicated in
288 GB HBM3E):

  - every rank holds the full graph + labels + block weights on its GPU;
  - each chunk's position range is split evenly across ranks; rank r runs
    phase A (gain/select) only for its slice;
  - proposal lists are all-gathered (RCCL over xGMI; fixed-capacity buffers
    + a count vector per chunk);
  - EVERY rank runs the identical deterministic commit on the concatenated
    list, so labels/weights stay bit-identical across ranks with no further
    broadcast -- multi-GPU results equal single-GPU results exactly.

The collective layer is injectable for CPU (gloo) testing of the sharding
logic; compute still requires a GPU.
"""

import numpy as np


def pos_count(n):
    """Position space of the 64-vertex-unit permutation (lp_common.h)."""
    return ((n + 63) // 64) * 64


def chunk_size(n, num_chunks=64):
    nu = (n + 63) // 64
    return ((nu + num_chunks - 1) // num_chunks) * 64


def chunk_ranges(n, num_chunks, chunk):
    C = chunk_size(n, num_chunks)
    lo = chunk * C
    hi = min(lo + C, pos_count(n))
    return lo, hi


def rank_slice(lo, hi, rank, world):
    span = hi - lo
    return lo + (rank * span) // world, lo + ((rank + 1) * span) // world


def refine_dist(eng, k, max_block_weights, partition, seed, iters, comm):
    """Sharded deterministic LP refinement.

    eng:  kaminpar_amd.LpEngine (or a test double with the same phase API)
    comm: object with fields rank, world and methods
          alloc_prop_buffer(cap) -> (buffer, data_ptr) and
          all_gather_props(buffer, count) -> (concatenated_ptr, total_count)
    Returns (cut, partition, stats).
    """
    n = len(partition)
    num_chunks = eng.num_chunks()
    C = chunk_size(n, num_chunks)
    cap = C // comm.world + 2

    eng.refine_begin(k, max_block_weights, partition, seed)
    buf, buf_ptr = comm.alloc_prop_buffer(cap)

    for it in range(iters):
        sweep_moves = 0
        for chunk in range(num_chunks):
            lo, hi = chunk_ranges(n, num_chunks, chunk)
            if lo >= hi:
                continue
            slo, shi = rank_slice(lo, hi, comm.rank, comm.world)
            cnt = eng.phase_a(it, chunk, slo, shi, buf_ptr, cap)
            cat_ptr, total = comm.all_gather_props(buf, cnt)
            sweep_moves += eng.commit(it, chunk, cat_ptr, total)
        if sweep_moves == 0:
            break

    return eng.refine_end()


def target_range(k, rank, world):
    """Rank's owned block range for the sharded commit."""
    return (rank * k) // world, ((rank + 1) * k) // world


def refine_dist_sharded(eng, k, max_block_weights, partition, seed, iters,
                        comm):
    """Sharded deterministic LP refinement with a SHARDED commit: each rank
    sorts and fixpoints only its own target-block range; per fixpoint round
    a k-sized de-admission delta is allreduced (exactly the block-weight
    reconciliation of kaminpar-dist/refinement/lp/lp_refiner.cc:296-333) and
    the final per-target rank-cutoffs are exchanged, so every rank applies
    the identical admitted set. Bit-identical to refine_dist / single-GPU.

    comm additionally provides zeros(shape, dtype) -> tensor and
    allreduce_(tensor) (in-place sum)."""
    n = len(partition)
    num_chunks = eng.num_chunks()
    C = chunk_size(n, num_chunks)
    cap = C // comm.world + 2
    c_lo, c_hi = target_range(k, comm.rank, comm.world)

    eng.refine_begin(k, max_block_weights, partition, seed)
    # share torch's stream with the engine: the collectives and the kernels
    # then order naturally and the per-call cross-stream syncs disappear
    stream_ptr = comm.engine_stream_ptr()
    if stream_ptr is not None:
        eng.set_stream(stream_ptr)
    buf, buf_ptr = comm.alloc_prop_buffer(cap)
    dep = comm.zeros(k + 1, "i64")
    delta = comm.zeros(k + 1, "i64")
    cutoff = comm.zeros(k, "i64")  # i64 cells (sum-as-allgather)
    arr = comm.zeros(k, "i64")
    moves_prev = 0

    for it in range(iters):
        for chunk in range(num_chunks):
            lo, hi = chunk_ranges(n, num_chunks, chunk)
            if lo >= hi:
                continue
            slo, shi = rank_slice(lo, hi, comm.rank, comm.world)
            cnt = eng.phase_a(it, chunk, slo, shi, buf_ptr, cap)
            cat_ptr, total = comm.all_gather_props(buf, cnt)

            dep.zero_()
            eng.shard_begin(c_lo, c_hi, cat_ptr, total, dep.data_ptr())
            comm.allreduce_(dep)  # global full-admission departures
            while True:
                eng.shard_round(c_lo, c_hi, dep.data_ptr(), delta.data_ptr())
                comm.allreduce_(delta)
                if int(delta[k].item()) == 0:
                    break
                dep[:k] -= delta[:k]
            cutoff.zero_()
            arr.zero_()
            eng.shard_finish_meta(c_lo, c_hi, cutoff.data_ptr(),
                                  arr.data_ptr())
            comm.allreduce_(cutoff)  # zeros elsewhere: sum == allgather
            comm.allreduce_(arr)
            comm.sync()
            eng.shard_apply(it, chunk, cat_ptr, total,
                            cutoff.data_ptr(), arr.data_ptr(),
                            dep.data_ptr())
        # one move-counter readback per sweep (not per chunk)
        moves_now = eng.get_stats().moves
        if moves_now == moves_prev:
            break
        moves_prev = moves_now

    return eng.refine_end()


def nccl_cpp_comm(rank, world):
    """Bootstrap an RCCL communicator for the C++ dist driver: rank 0
    generates the 128-byte unique id, torch.distributed broadcasts it,
    every rank inits. Returns an opaque handle (destroy with
    kaminpar_amd._lib.kmp_nccl_comm_destroy) -- None at world 1."""
    if world == 1:
        return None
    import ctypes

    import torch
    import torch.distributed as dist

    import kaminpar_amd as ka

    buf = (ctypes.c_char * 128)()
    if rank == 0:
        if ka._lib.kmp_nccl_unique_id(buf) != 0:
            raise RuntimeError("kmp_nccl_unique_id failed")
    t = torch.frombuffer(bytearray(buf.raw), dtype=torch.uint8).clone()
    dist.broadcast(t, src=0)
    buf = (ctypes.c_char * 128).from_buffer_copy(t.numpy().tobytes())
    comm = ka._lib.kmp_nccl_comm_init(world, rank, buf)
    if not comm:
        raise RuntimeError("kmp_nccl_comm_init failed")
    return comm


def refine_dist_cpp(eng, k, max_block_weights, partition, seed, iters,
                    rank, world, nccl_comm):
    """C++ RCCL-driven sharded refinement (kmp_lp_refine_dist): the whole
    chunk loop and all collectives run below the C-ABI; python only
    bootstraps the communicator."""
    return eng.refine_dist_cpp(k, max_block_weights, partition, seed, iters,
                               nccl_comm, rank, world)


class TorchComm:
    """torch.distributed-backed collective layer (nccl=RCCL on GPU)."""

    def __init__(self, device):
        import torch
        import torch.distributed as dist

        self.torch = torch
        self.dist = dist
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device
        self._cnt = torch.zeros(self.world, dtype=torch.int64, device=device)
        self._gather_bufs = None

    def alloc_prop_buffer(self, cap):
        # one extra trailing row carries the rank's proposal count, so a
        # single all_gather per chunk moves both payload and counts (the
        # collectives are per-link latency-bound over xGMI; halving the
        # count removes ~64 x sweeps collective round-trips per run)
        t = self.torch.zeros((cap + 1, 4), dtype=self.torch.int32,
                             device=self.device)
        self._gather_bufs = [
            self.torch.zeros((cap + 1, 4), dtype=self.torch.int32,
                             device=self.device)
            for _ in range(self.world)
        ]
        self._cap = cap
        self._buf = t
        return t, t.data_ptr()

    def zeros(self, size, dtype):
        assert dtype == "i64"
        return self.torch.zeros(size, dtype=self.torch.int64,
                                device=self.device)

    def engine_stream_ptr(self):
        if str(self.device).startswith("cuda"):
            return self.torch.cuda.current_stream().cuda_stream
        return None

    def allreduce_(self, t):
        self.dist.all_reduce(t)

    def sync(self):
        if str(self.device).startswith("cuda"):
            self.torch.cuda.current_stream().synchronize()

    def all_gather_props(self, buf, count):
        torch, dist = self.torch, self.dist
        buf[self._cap, 0] = count
        dist.all_gather(self._gather_bufs, buf)
        # ONE host sync for all ranks' counts (per-buffer .item() calls are
        # world separate syncs)
        counts = torch.stack(
            [b[self._cap, 0] for b in self._gather_bufs]).cpu().tolist()
        total = sum(counts)
        if total == 0:
            return buf.data_ptr(), 0
        cat = torch.cat([self._gather_bufs[r][: counts[r]] for r in range(self.world)])
        cat = cat.contiguous()
        self._cat_keepalive = cat  # keep device memory alive through commit
        # the engine commits on its own HIP stream: make sure the collective
        # + cat (torch's stream) are complete before handing over the pointer
        if cat.is_cuda:
            torch.cuda.current_stream().synchronize()
        return cat.data_ptr(), total


class LocalComm:
    """Single-process stand-in (world=1): no collectives."""

    rank = 0
    world = 1

    def __init__(self, torch_device=None):
        import torch

        self.torch = torch
        self.device = torch_device or "cuda:0"

    def alloc_prop_buffer(self, cap):
        t = self.torch.zeros((cap, 4), dtype=self.torch.int32, device=self.device)
        self._buf = t
        return t, t.data_ptr()

    def all_gather_props(self, buf, count):
        return buf.data_ptr(), count

    def zeros(self, size, dtype):
        assert dtype == "i64"
        return self.torch.zeros(size, dtype=self.torch.int64,
                                device=self.device)

    def engine_stream_ptr(self):
        if str(self.device).startswith("cuda"):
            return self.torch.cuda.current_stream().cuda_stream
        return None

    def allreduce_(self, t):
        pass

    def sync(self):
        if str(self.device).startswith("cuda"):
            self.torch.cuda.current_stream().synchronize()
