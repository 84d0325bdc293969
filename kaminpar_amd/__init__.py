"""kaminpar_amd: MI355X-native KaMinPar label-propagation hot path.

ctypes bindings over libkaminpar_lp.so (C ABI: include/kaminpar_lp.h).
Host-side graph handling works everywhere; the LP engine requires a GPU and
fails loudly if the HIP device or the native library is missing -- there is
no CPU fallback on the product path (the CPU oracle under oracle/ is test
infrastructure only).

Interop note: PyTorch wheels bundle their own HIP runtime; this module
preloads it (when torch is installed) before loading the native library so
the whole process shares one runtime regardless of import/initialization
order.
"""

import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libkaminpar_lp.so")


def _preload_torch_hip_runtime():
    """Bind everything to ONE HIP runtime.

    PyTorch wheels bundle their own libamdhip64 and load it via absolute
    RPATH paths, ignoring an already-loaded system copy -- two HSA clients in
    one process leave whichever initializes second unable to see the GPU.
    Preloading torch's copy under its SONAME makes this library resolve to
    it, and torch's later absolute-path load maps the same file. Without
    torch installed this is a no-op and the system runtime is used.
    """
    try:
        import importlib.util

        spec = importlib.util.find_spec("torch")
        if spec and spec.origin:
            cand = os.path.join(os.path.dirname(spec.origin), "lib", "libamdhip64.so")
            if os.path.exists(cand):
                ctypes.CDLL(cand, mode=ctypes.RTLD_GLOBAL)
    except Exception:
        pass  # fall back to the system runtime


def _load():
    if not os.path.exists(_LIB_PATH):
        raise ImportError(
            f"kaminpar_amd: native library not found at {_LIB_PATH}; "
            "build it with __graft_entry__.build() or kaminpar_amd/csrc/build.sh"
        )
    _preload_torch_hip_runtime()
    lib = ctypes.CDLL(_LIB_PATH)

    u32, u64, i32, i64 = ctypes.c_uint32, ctypes.c_uint64, ctypes.c_int32, ctypes.c_int64
    p = ctypes.POINTER
    vp = ctypes.c_void_p

    lib.kmp_graph_from_csr.restype = vp
    lib.kmp_graph_from_csr.argtypes = [u32, u64, p(u32), p(u32), p(i32), p(i32)]
    lib.kmp_graph_from_csr64.restype = vp
    lib.kmp_graph_from_csr64.argtypes = [u32, u64, p(u64), p(u32), p(i32), p(i32)]
    lib.kmp_gen_rmat.restype = vp
    lib.kmp_gen_rmat.argtypes = [ctypes.c_int, ctypes.c_int, u64]
    lib.kmp_gen_rgg2d.restype = vp
    lib.kmp_gen_rgg2d.argtypes = [u32, ctypes.c_double, u64]
    lib.kmp_read_metis.restype = vp
    lib.kmp_read_metis.argtypes = [ctypes.c_char_p]
    lib.kmp_write_metis.restype = ctypes.c_int
    lib.kmp_write_metis.argtypes = [vp, ctypes.c_char_p]
    lib.kmp_read_parhip.restype = vp
    lib.kmp_read_parhip.argtypes = [ctypes.c_char_p]
    lib.kmp_write_parhip.restype = ctypes.c_int
    lib.kmp_write_parhip.argtypes = [vp, ctypes.c_char_p]
    lib.kmp_graph_n.restype = u32
    lib.kmp_graph_n.argtypes = [vp]
    lib.kmp_graph_m.restype = u64
    lib.kmp_graph_m.argtypes = [vp]
    lib.kmp_graph_xadj.restype = p(u32)
    lib.kmp_graph_xadj.argtypes = [vp]
    lib.kmp_graph_xadj64.restype = p(u64)
    lib.kmp_graph_xadj64.argtypes = [vp]
    lib.kmp_graph_adjncy.restype = p(u32)
    lib.kmp_graph_adjncy.argtypes = [vp]
    lib.kmp_graph_vwgt.restype = p(i32)
    lib.kmp_graph_vwgt.argtypes = [vp]
    lib.kmp_graph_adjwgt.restype = p(i32)
    lib.kmp_graph_adjwgt.argtypes = [vp]
    lib.kmp_graph_total_node_weight.restype = i64
    lib.kmp_graph_total_node_weight.argtypes = [vp]
    lib.kmp_graph_free.argtypes = [vp]
    lib.kmp_edge_cut_host.restype = i64
    lib.kmp_edge_cut_host.argtypes = [vp, p(u32)]
    lib.kmp_max_block_weight.restype = i64
    lib.kmp_max_block_weight.argtypes = [vp, u32, ctypes.c_double]
    lib.kmp_rearrange_degree_buckets.restype = vp
    lib.kmp_rearrange_degree_buckets.argtypes = [vp, p(u32)]
    lib.kmp_initial_partition.restype = ctypes.c_int
    lib.kmp_initial_partition.argtypes = [vp, u32, i64, ctypes.c_int, p(u32)]
    lib.kmp_balance_partition.restype = ctypes.c_int
    lib.kmp_balance_partition.argtypes = [vp, u32, i64, p(u32)]
    lib.kmp_kway_fm.restype = ctypes.c_int
    lib.kmp_kway_fm.argtypes = [vp, u32, p(i64), p(u32),
                                ctypes.c_int, ctypes.c_int]
    lib.kmp_bisect_subset.restype = ctypes.c_int
    lib.kmp_bisect_subset.argtypes = [vp, p(u32), u32, i64, i64, i64,
                                      ctypes.c_int, p(ctypes.c_uint8)]
    lib.kmp_bisect_subset_ml.restype = ctypes.c_int
    lib.kmp_bisect_subset_ml.argtypes = [vp, p(u32), u32, i64, i64, i64,
                                         ctypes.c_int, p(ctypes.c_uint8)]
    lib.kmp_bisect_subset_fast.restype = ctypes.c_int
    lib.kmp_bisect_subset_fast.argtypes = [vp, p(u32), u32, i64, i64, i64,
                                           ctypes.c_int, p(ctypes.c_uint8)]
    lib.kmp_partition.restype = i64
    lib.kmp_partition.argtypes = [vp, u32, ctypes.c_double, u64, ctypes.c_int,
                                  u32, u32, ctypes.c_int, p(u32)]
    lib.kmp_partition_deep.restype = i64
    lib.kmp_partition_deep.argtypes = [vp, u32, ctypes.c_double, u64,
                                       ctypes.c_int, u32, u32, u32,
                                       ctypes.c_int, p(u32)]

    lib.kmp_lp_create.restype = vp
    lib.kmp_lp_create.argtypes = [vp]
    lib.kmp_lp_free.argtypes = [vp]
    lib.kmp_lp_refine.restype = i64
    lib.kmp_lp_refine.argtypes = [vp, u32, p(i64), p(u32), u64, ctypes.c_int, vp]
    lib.kmp_lp_balance.restype = i64
    lib.kmp_lp_balance.argtypes = [vp, u32, p(i64), p(u32), u64, ctypes.c_int, vp]
    lib.kmp_lp_set_communities.restype = ctypes.c_int
    lib.kmp_lp_set_communities.argtypes = [vp, p(u32)]
    lib.kmp_lp_rearrange_degree_buckets.restype = ctypes.c_int
    lib.kmp_lp_rearrange_degree_buckets.argtypes = [vp, p(u32)]
    lib.kmp_lp_underload.restype = i64
    lib.kmp_lp_underload.argtypes = [vp, u32, p(i64), p(i64), p(u32), u64,
                                     ctypes.c_int, vp]
    lib.kmp_lp_cluster.restype = i64
    lib.kmp_lp_cluster.argtypes = [vp, i64, u32, p(u32), u64, ctypes.c_int, vp]
    lib.kmp_lp_num_chunks.restype = u32
    lib.kmp_lp_num_chunks.argtypes = [vp]
    lib.kmp_lp_refine_begin.restype = ctypes.c_int
    lib.kmp_lp_refine_begin.argtypes = [vp, u32, p(i64), p(u32), u64]
    lib.kmp_lp_phase_a.restype = i64
    lib.kmp_lp_phase_a.argtypes = [vp, ctypes.c_int, u32, u32, u32, vp, u32]
    lib.kmp_lp_commit.restype = i64
    lib.kmp_lp_commit.argtypes = [vp, ctypes.c_int, u32, vp, u32]
    lib.kmp_nccl_unique_id.restype = ctypes.c_int
    lib.kmp_nccl_unique_id.argtypes = [vp]
    lib.kmp_nccl_comm_init.restype = vp
    lib.kmp_nccl_comm_init.argtypes = [ctypes.c_int, ctypes.c_int, vp]
    lib.kmp_nccl_comm_destroy.restype = None
    lib.kmp_nccl_comm_destroy.argtypes = [vp]
    lib.kmp_lp_refine_dist.restype = i64
    lib.kmp_lp_refine_dist.argtypes = [vp, u32, p(i64), p(u32), u64,
                                       ctypes.c_int, vp, ctypes.c_int,
                                       ctypes.c_int, vp]
    lib.kmp_lp_set_stream.restype = ctypes.c_int
    lib.kmp_lp_set_stream.argtypes = [vp, vp]
    lib.kmp_lp_shard_begin.restype = ctypes.c_int
    lib.kmp_lp_shard_begin.argtypes = [vp, u32, u32, vp, u32, vp]
    lib.kmp_lp_shard_round.restype = ctypes.c_int
    lib.kmp_lp_shard_round.argtypes = [vp, u32, u32, vp, vp]
    lib.kmp_lp_shard_finish_meta.restype = ctypes.c_int
    lib.kmp_lp_shard_finish_meta.argtypes = [vp, u32, u32, vp, vp]
    lib.kmp_lp_shard_apply.restype = i64
    lib.kmp_lp_shard_apply.argtypes = [vp, ctypes.c_int, u32, vp, u32, vp, vp, vp]
    lib.kmp_lp_refine_end.restype = i64
    lib.kmp_lp_refine_end.argtypes = [vp, p(u32), vp]
    lib.kmp_lp_reset.restype = ctypes.c_int
    lib.kmp_lp_reset.argtypes = [vp]
    lib.kmp_lp_run_sweeps.restype = i64
    lib.kmp_lp_run_sweeps.argtypes = [vp, ctypes.c_int]
    lib.kmp_lp_get_stats.restype = ctypes.c_int
    lib.kmp_lp_get_stats.argtypes = [vp, vp]
    lib.kmp_contract.restype = i64
    lib.kmp_contract.argtypes = [vp, p(u32), p(u32), ctypes.POINTER(vp)]
    lib.kmp_contract_engine.restype = i64
    lib.kmp_contract_engine.argtypes = [vp, p(u32), p(u32), ctypes.POINTER(vp)]
    lib.kmp_lp_download_graph.restype = vp
    lib.kmp_lp_download_graph.argtypes = [vp]
    lib.kmp_lp_n.restype = u32
    lib.kmp_lp_n.argtypes = [vp]
    lib.kmp_lp_m.restype = ctypes.c_uint64
    lib.kmp_lp_m.argtypes = [vp]
    return lib


_lib = _load()


class Stats(ctypes.Structure):
    _fields_ = [
        ("arcs_scanned", ctypes.c_uint64),
        ("moves", ctypes.c_uint64),
        ("phase_a_ns", ctypes.c_uint64),
        ("total_ns", ctypes.c_uint64),
        ("num_clusters", ctypes.c_uint64),
        ("edge_cut", ctypes.c_int64),
    ]


def _u32p(a):
    assert a.dtype == np.uint32 and a.flags["C_CONTIGUOUS"]
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_uint32))


def _i64p(a):
    assert a.dtype == np.int64 and a.flags["C_CONTIGUOUS"]
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


class Graph:
    """Host CSR graph in the reference layout (csr_graph.h:27-33)."""

    def __init__(self, handle):
        if not handle:
            raise ValueError("graph construction failed")
        self._h = handle

    @classmethod
    def from_csr(cls, xadj, adjncy, vwgt=None, adjwgt=None):
        adjncy = np.ascontiguousarray(adjncy, dtype=np.uint32)
        n = len(xadj) - 1
        m = len(adjncy)
        vp = None
        ap = None
        if vwgt is not None:
            vwgt = np.ascontiguousarray(vwgt, dtype=np.int32)
            vp = vwgt.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))
        if adjwgt is not None:
            adjwgt = np.ascontiguousarray(adjwgt, dtype=np.int32)
            ap = adjwgt.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))
        if m >= (1 << 32):
            # EdgeID-64 path (device offsets are 64-bit either way)
            xadj = np.ascontiguousarray(xadj, dtype=np.uint64)
            xp = xadj.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
            return cls(_lib.kmp_graph_from_csr64(n, m, xp, _u32p(adjncy), vp, ap))
        xadj = np.ascontiguousarray(xadj, dtype=np.uint32)
        return cls(_lib.kmp_graph_from_csr(n, m, _u32p(xadj), _u32p(adjncy), vp, ap))

    @classmethod
    def rmat(cls, scale, edgefactor=8, seed=42):
        return cls(_lib.kmp_gen_rmat(scale, edgefactor, seed))

    @classmethod
    def rgg2d(cls, n, avg_deg=16.0, seed=42):
        return cls(_lib.kmp_gen_rgg2d(n, avg_deg, seed))

    @classmethod
    def read_metis(cls, path):
        return cls(_lib.kmp_read_metis(os.fsencode(path)))

    @classmethod
    def read_parhip(cls, path):
        return cls(_lib.kmp_read_parhip(os.fsencode(path)))

    def write_metis(self, path):
        if _lib.kmp_write_metis(self._h, os.fsencode(path)) != 0:
            raise IOError(f"cannot write {path}")

    def write_parhip(self, path):
        if _lib.kmp_write_parhip(self._h, os.fsencode(path)) != 0:
            raise IOError(f"cannot write {path}")

    @property
    def n(self):
        return _lib.kmp_graph_n(self._h)

    @property
    def m(self):
        return _lib.kmp_graph_m(self._h)

    @property
    def xadj(self):
        p = _lib.kmp_graph_xadj(self._h)
        if not p:
            # 64-bit-offset graph (m >= 2^32): expose the wide offsets
            p64 = _lib.kmp_graph_xadj64(self._h)
            return np.ctypeslib.as_array(p64, shape=(self.n + 1,))
        return np.ctypeslib.as_array(p, shape=(self.n + 1,))

    @property
    def adjncy(self):
        if self.m == 0:
            return np.zeros(0, dtype=np.uint32)  # null data ptr on empty CSR
        return np.ctypeslib.as_array(_lib.kmp_graph_adjncy(self._h), shape=(self.m,))

    @property
    def total_node_weight(self):
        return _lib.kmp_graph_total_node_weight(self._h)

    def edge_cut(self, labels):
        labels = np.ascontiguousarray(labels, dtype=np.uint32)
        return _lib.kmp_edge_cut_host(self._h, _u32p(labels))

    def max_block_weight(self, k, eps=0.03):
        return _lib.kmp_max_block_weight(self._h, k, eps)

    def partition_native(self, k, eps=0.03, seed=1, iters=5):
        """Full multilevel partition driven entirely from the C ABI
        (kmp_partition; the shape of KaMinPar::compute_partition).
        Bit-identical to kaminpar_amd.partition.partition. Requires a GPU.

        Returns (cut, partition)."""
        part = np.zeros(self.n, dtype=np.uint32)
        cut = _lib.kmp_partition(self._h, k, eps, seed, iters, 0, 0, 0,
                                 _u32p(part))
        if cut < 0:
            raise RuntimeError("kmp_partition failed")
        return cut, part

    def balance_partition(self, k, cap, part):
        """Gain-aware overload balancer (uniform cap), in place."""
        part = np.ascontiguousarray(part, dtype=np.uint32)
        _lib.kmp_balance_partition(self._h, k, int(cap), _u32p(part))
        return part

    def kway_fm(self, k, caps, part, max_passes=0, max_fruitless=0):
        """Deterministic k-way boundary FM (best-prefix rollback), in
        place; caps is a k-length per-block hard cap array (0 closes a
        block)."""
        caps = np.ascontiguousarray(caps, dtype=np.int64)
        part = np.ascontiguousarray(part, dtype=np.uint32)
        _lib.kmp_kway_fm(self._h, k,
                         caps.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
                         _u32p(part), max_passes, max_fruitless)
        return part

    def bisect_subset(self, nodes, target1, cap1, cap2, reps=8):
        """Bisect an arbitrary vertex subset (greedy grow + FM, best of
        reps). Returns a boolean side array aligned with `nodes`."""
        nodes = np.ascontiguousarray(nodes, dtype=np.uint32)
        side = np.zeros(len(nodes), dtype=np.uint8)
        rc = _lib.kmp_bisect_subset(
            self._h, _u32p(nodes), len(nodes), int(target1), int(cap1),
            int(cap2), reps, side.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
        if rc != 0:
            raise RuntimeError("kmp_bisect_subset failed")
        return side.astype(bool)

    def bisect_subset_fast(self, nodes, target1, cap1, cap2, reps=8):
        """O(m log n) bisection (lazy-PQ greedy grow + FM); for subgraphs
        beyond a few thousand vertices."""
        nodes = np.ascontiguousarray(nodes, dtype=np.uint32)
        side = np.zeros(len(nodes), dtype=np.uint8)
        rc = _lib.kmp_bisect_subset_fast(
            self._h, _u32p(nodes), len(nodes), int(target1), int(cap1),
            int(cap2), reps, side.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
        if rc != 0:
            raise RuntimeError("kmp_bisect_subset_fast failed")
        return side.astype(bool)

    def bisect_subset_ml(self, nodes, target1, cap1, cap2, reps=8):
        """Multilevel bisection of a vertex subset (heavy-edge matching
        + FM at every level; best of reps)."""
        nodes = np.ascontiguousarray(nodes, dtype=np.uint32)
        side = np.zeros(len(nodes), dtype=np.uint8)
        rc = _lib.kmp_bisect_subset_ml(
            self._h, _u32p(nodes), len(nodes), int(target1), int(cap1),
            int(cap2), reps, side.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
        if rc != 0:
            raise RuntimeError("kmp_bisect_subset_ml failed")
        return side.astype(bool)

    def partition_deep_native(self, k, eps=0.03, seed=1, iters=5):
        """Progressive-k multilevel partition driven entirely from the C
        ABI (kmp_partition_deep). Bit-identical to
        kaminpar_amd.partition.partition_deep. Requires a GPU."""
        part = np.zeros(self.n, dtype=np.uint32)
        cut = _lib.kmp_partition_deep(self._h, k, eps, seed, iters, 0, 0, 0,
                                      0, _u32p(part))
        if cut < 0:
            raise RuntimeError("kmp_partition_deep failed")
        return cut, part

    def initial_partition_native(self, k, max_block_weight, reps=8):
        """C++ recursive-bisection initial partitioning (equivalent to
        kaminpar_amd.partition.initial_partition; host-only, no GPU)."""
        part = np.zeros(self.n, dtype=np.uint32)
        rc = _lib.kmp_initial_partition(self._h, k, max_block_weight, reps,
                                        _u32p(part))
        if rc != 0:
            raise RuntimeError("kmp_initial_partition failed")
        return part

    def rearrange_degree_buckets(self):
        """Degree-bucket rearrangement (the reference's default
        NodeOrdering::DEGREE_BUCKETS preprocessing).

        Returns (permuted_graph, perm) with perm[u_old] = u_new; a labelling
        l_new on the permuted graph maps back as l_old = l_new[perm]."""
        perm = np.zeros(self.n, dtype=np.uint32)
        h = _lib.kmp_rearrange_degree_buckets(self._h, _u32p(perm))
        return Graph(h), perm

    def __del__(self):
        h = getattr(self, "_h", None)
        if h and _lib is not None:
            _lib.kmp_graph_free(h)
            self._h = None


def random_partition(n, k, seed=42):
    """Deterministic pseudo-random balanced-in-expectation partition."""
    with np.errstate(over="ignore"):
        u = np.arange(n, dtype=np.uint64)
        x = u + np.uint64((0x9E3779B97F4A7C15 * (seed + 1)) & 0xFFFFFFFFFFFFFFFF)
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        x = x ^ (x >> np.uint64(31))
    return (x % np.uint64(k)).astype(np.uint32)


class LpEngine:
    """Device-resident LP engine (requires a GPU; fails loudly otherwise)."""

    def __init__(self, graph=None, _handle=None):
        self._graph = graph  # keep alive (None for device-resident engines)
        if _handle is not None:
            self._h = _handle
        else:
            self._h = _lib.kmp_lp_create(graph._h)
        if not self._h:
            raise RuntimeError(
                "kaminpar_amd: LP engine creation failed (no HIP GPU available?)"
            )
        self.n = int(_lib.kmp_lp_n(self._h))
        self.m = int(_lib.kmp_lp_m(self._h))

    def refine(self, k, max_block_weights, partition, seed=1, iters=5):
        """Deterministic LP refinement; returns (cut, partition, Stats)."""
        part = np.ascontiguousarray(partition, dtype=np.uint32).copy()
        mbw = np.ascontiguousarray(max_block_weights, dtype=np.int64)
        stats = Stats()
        cut = _lib.kmp_lp_refine(
            self._h, k, _i64p(mbw), _u32p(part), seed, iters, ctypes.byref(stats)
        )
        if cut < 0:
            raise RuntimeError("kmp_lp_refine failed")
        return cut, part, stats

    def balance(self, k, max_block_weights, partition, seed=1, iters=5):
        """Overload-balancer mode: repair an infeasible partition (blocks
        over their caps shed boundary vertices at best admissible gain).
        Returns (cut, partition, Stats)."""
        part = np.ascontiguousarray(partition, dtype=np.uint32).copy()
        mbw = np.ascontiguousarray(max_block_weights, dtype=np.int64)
        stats = Stats()
        cut = _lib.kmp_lp_balance(
            self._h, k, _i64p(mbw), _u32p(part), seed, iters, ctypes.byref(stats)
        )
        if cut < 0:
            raise RuntimeError("kmp_lp_balance failed")
        return cut, part, stats

    def underload(self, k, max_block_weights, min_block_weights, partition,
                  seed=1, iters=5):
        """Underload-balancer mode: fill blocks below their minimum weight
        (presets.cc:332-338 UNDERLOAD_BALANCER role; semantics restated from
        refinement/balancer/underload_balancer.cc). Returns
        (cut, partition, Stats)."""
        part = np.ascontiguousarray(partition, dtype=np.uint32).copy()
        mbw = np.ascontiguousarray(max_block_weights, dtype=np.int64)
        mnw = np.ascontiguousarray(min_block_weights, dtype=np.int64)
        stats = Stats()
        cut = _lib.kmp_lp_underload(
            self._h, k, _i64p(mbw), _i64p(mnw), _u32p(part), seed, iters,
            ctypes.byref(stats)
        )
        if cut < 0:
            raise RuntimeError("kmp_lp_underload failed")
        return cut, part, stats

    def rearrange_degree_buckets(self):
        """On-GPU degree-bucket rearrangement of the engine's CSR (the
        reference's default preprocessing, permutator.cc:36-110).
        Returns perm with perm[u_old] = u_new."""
        perm = np.zeros(self.n, dtype=np.uint32)
        rc = _lib.kmp_lp_rearrange_degree_buckets(self._h, _u32p(perm))
        if rc != 0:
            raise RuntimeError("kmp_lp_rearrange_degree_buckets failed")
        return perm

    def set_communities(self, communities):
        """Restrict clustering merges to stay within communities
        (coarsening/clusterer.h:35, lp_clusterer.cc:193-194). None clears."""
        if communities is None:
            _lib.kmp_lp_set_communities(self._h, None)
            self._comm_keepalive = None
            return
        comm = np.ascontiguousarray(communities, dtype=np.uint32)
        _lib.kmp_lp_set_communities(self._h, _u32p(comm))
        self._comm_keepalive = comm

    def cluster(self, max_cluster_weight, clustering=None, desired=0, seed=1, iters=5):
        """Deterministic LP clustering; returns (n_clusters, clustering, Stats)."""
        clus = np.zeros(self.n, dtype=np.uint32)
        stats = Stats()
        nc = _lib.kmp_lp_cluster(
            self._h, max_cluster_weight, desired, _u32p(clus), seed, iters,
            ctypes.byref(stats),
        )
        if nc < 0:
            raise RuntimeError("kmp_lp_cluster failed")
        return nc, clus, stats

    # sharded API (multi-GPU orchestration; see kaminpar_amd/multi.py)
    def num_chunks(self):
        return _lib.kmp_lp_num_chunks(self._h)

    def refine_begin(self, k, max_block_weights, partition, seed=1):
        part = np.ascontiguousarray(partition, dtype=np.uint32)
        mbw = np.ascontiguousarray(max_block_weights, dtype=np.int64)
        rc = _lib.kmp_lp_refine_begin(self._h, k, _i64p(mbw), _u32p(part), seed)
        if rc != 0:
            raise RuntimeError("kmp_lp_refine_begin failed")

    def phase_a(self, it, chunk, pos_lo, pos_hi, d_out_ptr, cap):
        cnt = _lib.kmp_lp_phase_a(self._h, it, chunk, pos_lo, pos_hi, d_out_ptr, cap)
        if cnt < 0:
            raise RuntimeError("kmp_lp_phase_a failed")
        return int(cnt)

    def commit(self, it, chunk, d_props_ptr, count):
        mv = _lib.kmp_lp_commit(self._h, it, chunk, d_props_ptr, count)
        if mv < 0:
            raise RuntimeError("kmp_lp_commit failed")
        return int(mv)

    def refine_dist_cpp(self, k, max_block_weights, partition, seed, iters,
                        nccl_comm, rank, world):
        """C++ RCCL-driven sharded refinement (see kmp_lp_refine_dist)."""
        part = np.ascontiguousarray(partition, dtype=np.uint32).copy()
        mbw = np.ascontiguousarray(max_block_weights, dtype=np.int64)
        stats = Stats()
        cut = _lib.kmp_lp_refine_dist(
            self._h, k, _i64p(mbw), _u32p(part), seed, iters,
            nccl_comm, rank, world, ctypes.byref(stats))
        if cut < 0:
            raise RuntimeError("kmp_lp_refine_dist failed")
        return cut, part, stats

    def set_stream(self, stream_ptr):
        """Adopt an external HIP stream (torch: cuda.current_stream().cuda_stream);
        None restores the engine's own stream."""
        rc = _lib.kmp_lp_set_stream(self._h, stream_ptr)
        if rc != 0:
            raise RuntimeError("kmp_lp_set_stream failed")

    def shard_begin(self, c_lo, c_hi, d_props_ptr, count, d_dep_out_ptr):
        """Sharded commit step 1: sort own targets from the all-gathered
        list + full-admission departure contributions (device ptrs)."""
        rc = _lib.kmp_lp_shard_begin(self._h, c_lo, c_hi, d_props_ptr, count,
                                     d_dep_out_ptr)
        if rc != 0:
            raise RuntimeError("kmp_lp_shard_begin failed")

    def shard_round(self, c_lo, c_hi, d_dep_global_ptr, d_delta_out_ptr):
        rc = _lib.kmp_lp_shard_round(self._h, c_lo, c_hi, d_dep_global_ptr,
                                     d_delta_out_ptr)
        if rc != 0:
            raise RuntimeError("kmp_lp_shard_round failed")

    def shard_finish_meta(self, c_lo, c_hi, d_cutoff_ptr, d_arr_ptr):
        rc = _lib.kmp_lp_shard_finish_meta(self._h, c_lo, c_hi, d_cutoff_ptr,
                                           d_arr_ptr)
        if rc != 0:
            raise RuntimeError("kmp_lp_shard_finish_meta failed")

    def shard_apply(self, it, chunk, d_props_ptr, count, d_cutoff_ptr,
                    d_arr_ptr, d_dep_ptr):
        mv = _lib.kmp_lp_shard_apply(self._h, it, chunk, d_props_ptr, count,
                                     d_cutoff_ptr, d_arr_ptr, d_dep_ptr)
        if mv < 0:
            raise RuntimeError("kmp_lp_shard_apply failed")
        return int(mv)

    def reset(self):
        if _lib.kmp_lp_reset(self._h) != 0:
            raise RuntimeError("kmp_lp_reset failed")

    def run_sweeps(self, iters=5):
        mv = _lib.kmp_lp_run_sweeps(self._h, iters)
        if mv < 0:
            raise RuntimeError("kmp_lp_run_sweeps failed")
        return int(mv)

    def get_stats(self):
        stats = Stats()
        _lib.kmp_lp_get_stats(self._h, ctypes.byref(stats))
        return stats

    def contract(self, clustering):
        """Contract a clustering into the coarse graph (GPU).

        Returns (coarse_graph: Graph, mapping: np.ndarray[u32])."""
        clus = np.ascontiguousarray(clustering, dtype=np.uint32)
        mapping = np.zeros(self.n, dtype=np.uint32)
        out = ctypes.c_void_p()
        c_n = _lib.kmp_contract(self._h, _u32p(clus), _u32p(mapping), ctypes.byref(out))
        if c_n < 0:
            raise RuntimeError("kmp_contract failed")
        return Graph(out.value), mapping

    def contract_engine(self, clustering):
        """Contract a clustering and hand the coarse graph directly to a NEW
        engine without a host round-trip (the coarse CSR stays in HBM).

        Returns (coarse_engine: LpEngine, mapping: np.ndarray[u32]). Results
        are identical to contract() + LpEngine(coarse_graph)."""
        clus = np.ascontiguousarray(clustering, dtype=np.uint32)
        mapping = np.zeros(self.n, dtype=np.uint32)
        out = ctypes.c_void_p()
        c_n = _lib.kmp_contract_engine(self._h, _u32p(clus), _u32p(mapping),
                                       ctypes.byref(out))
        if c_n < 0:
            raise RuntimeError("kmp_contract_engine failed")
        return LpEngine(_handle=out.value), mapping

    def download_graph(self):
        """Download the engine's device-resident CSR into a host Graph."""
        h = _lib.kmp_lp_download_graph(self._h)
        return Graph(h)

    def refine_end(self):
        part = np.zeros(self.n, dtype=np.uint32)
        stats = Stats()
        cut = _lib.kmp_lp_refine_end(self._h, _u32p(part), ctypes.byref(stats))
        return cut, part, stats

    def __del__(self):
        h = getattr(self, "_h", None)
        if h and _lib is not None:
            _lib.kmp_lp_free(h)
            self._h = None
