/*
 * C ABI of the MI355X-native KaMinPar label-propagation hot path.
 *
 * This is the drop-in boundary for the reference's LP operator seams:
 *   - kmp_lp_cluster_*  replaces Clusterer::compute_clustering behind
 *     kaminpar-shm/coarsening/clusterer.h:19-47 (LPClustering,
 *     lp_clusterer.cc:395) including set_max_cluster_weight /
 *     set_desired_cluster_count (clusterer.h:35-36);
 *   - kmp_lp_refine_*   replaces Refiner::refine behind
 *     kaminpar-shm/refinement/refiner.h:18-57 (LabelPropagationRefiner,
 *     lp_refiner.cc:370-372);
 *   - graph handles keep the reference CSR layout exactly
 *     (CSRGraphMemory, kaminpar-shm/datastructures/csr_graph.h:27-33:
 *     xadj[n+1], adjncy[m] with both arc directions, optional vwgt/adjwgt);
 *   - kmp_edge_cut mirrors metrics::edge_cut (kaminpar-shm/metrics.cc:37-59).
 *
 * Types follow include/kaminpar-shm/ckaminpar.h:27-52: NodeID/EdgeID u32
 * (u64 edge ids are a planned variant for >4G-arc graphs), NodeWeight/
 * EdgeWeight i32, BlockID u32, BlockWeight i64 here (the reference uses
 * NodeWeight-width block weights; i64 caps avoid overflow at scale-28).
 *
 * Plain pointers and sizes only; no torch types. Compute entry points
 * REQUIRE a GPU (they abort with a clear error if no HIP device is present);
 * graph generation/IO entry points are host-only.
 */
#ifndef KAMINPAR_LP_H
#define KAMINPAR_LP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct kmp_graph_t kmp_graph_t; /* host-resident CSR graph */
typedef struct kmp_lp_t kmp_lp_t;       /* device-resident LP engine */

/* ---------------------------------------------------------------- graphs */

/* Copy a CSR graph (reference layout, csr_graph.h:27-33). vwgt/adjwgt may be
 * NULL for unit weights. Returns NULL on invalid input. */
kmp_graph_t *kmp_graph_from_csr(
    uint32_t n,
    uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt
);

/* Graph500-parameter R-MAT generator (a=.57,b=.19,c=.19,d=.05), symmetrized
 * and deduplicated, unit weights; deterministic in (scale, edgefactor, seed).
 * n = 2^scale, #undirected edges before dedup = edgefactor * n. */
kmp_graph_t *kmp_gen_rmat(int scale, int edgefactor, uint64_t seed);

/* 2D random geometric graph on the unit square: n points, radius chosen for
 * the given average degree; deterministic in (n, avg_deg, seed). */
kmp_graph_t *kmp_gen_rgg2d(uint32_t n, double avg_deg, uint64_t seed);

/* METIS ASCII reader (kaminpar-io/metis_parser.h:17-25 format). */
kmp_graph_t *kmp_read_metis(const char *path);

/* METIS ASCII writer (counterpart of kmp_read_metis). */
int kmp_write_metis(const kmp_graph_t *g, const char *path);

/* ParHIP binary reader/writer (docs/graph_file_format.md "ParHIP Graph
 * File Format"; kaminpar-io/parhip_parser.cc:42-136): 24-byte header
 * (version bit-field, n, m), byte offsets, adjacency, optional weights.
 * Reader accepts 32- and 64-bit stored ids/weights that fit u32/i32;
 * writer emits 32-bit ids (64-bit offsets when the file is large). */
kmp_graph_t *kmp_read_parhip(const char *path);
int kmp_write_parhip(const kmp_graph_t *g, const char *path);

/* EdgeID-64 construction (ckaminpar.h:33-37 KAMINPAR_64BIT_EDGE_IDS
 * analogue): CSR with 64-bit offsets for graphs of >= 2^32 directed arcs.
 * Such graphs run on the LP engine (device offsets are always 64-bit);
 * host-side utilities (IO, rearrangement, host edge cut) require 32-bit. */
kmp_graph_t *kmp_graph_from_csr64(
    uint32_t n,
    uint64_t m,
    const uint64_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt
);
const uint64_t *kmp_graph_xadj64(const kmp_graph_t *g); /* null if 32-bit */

uint32_t kmp_graph_n(const kmp_graph_t *g);
uint64_t kmp_graph_m(const kmp_graph_t *g); /* number of directed arcs */
const uint32_t *kmp_graph_xadj(const kmp_graph_t *g);
const uint32_t *kmp_graph_adjncy(const kmp_graph_t *g);
const int32_t *kmp_graph_vwgt(const kmp_graph_t *g);     /* NULL if unit */
const int32_t *kmp_graph_adjwgt(const kmp_graph_t *g);   /* NULL if unit */
int64_t kmp_graph_total_node_weight(const kmp_graph_t *g);
void kmp_graph_free(kmp_graph_t *g);

/* Sequential host edge cut of a labelling (metrics.cc:73-84). */
int64_t kmp_edge_cut_host(const kmp_graph_t *g, const uint32_t *labels);

/* Degree-bucket rearrangement (the reference's default preprocessing,
 * NodeOrdering::DEGREE_BUCKETS: graphutils/permutator.h:30-128, stable
 * counting sort by floor_log2(deg)+1 with isolated vertices last).
 * perm_out[u_old] = u_new, n entries; returns the permuted graph. */
kmp_graph_t *kmp_rearrange_degree_buckets(const kmp_graph_t *g,
                                          uint32_t *perm_out);

/* Max block weight as the reference's PartitionContext computes it for
 * uniform epsilon (context.cc:27-39): (1+eps) * ceil(total_weight / k). */
int64_t kmp_max_block_weight(const kmp_graph_t *g, uint32_t k, double eps);

/* ------------------------------------------------------------- LP engine */

/* Run statistics, filled by the compute entry points. */
typedef struct kmp_lp_stats_t {
  uint64_t arcs_scanned;    /* directed arcs scanned over all sweeps */
  uint64_t moves;           /* committed moves */
  uint64_t phase_a_ns;      /* HIP-event time in gain/select kernels */
  uint64_t total_ns;        /* wall time of the LP region (device-synced) */
  uint64_t num_clusters;    /* clusterer: final non-empty clusters */
  int64_t edge_cut;         /* refiner: final cut (device-computed) */
} kmp_lp_stats_t;

/* Create an engine on the current HIP device and upload the graph.
 * Aborts (returns NULL + message on stderr) if no GPU is available. */
kmp_lp_t *kmp_lp_create(const kmp_graph_t *g);
void kmp_lp_free(kmp_lp_t *e);

/* Deterministic LP refinement (chunk-synchronous schedule; see
 * oracle/lp_oracle.cpp header for the schedule contract). partition: in/out,
 * n entries, values < k. max_block_weights: k entries. Returns the final
 * edge cut, or -1 on error. */
int64_t kmp_lp_refine(
    kmp_lp_t *e,
    uint32_t k,
    const int64_t *max_block_weights,
    uint32_t *partition,
    uint64_t seed,
    int iters,
    kmp_lp_stats_t *stats
);

/* Overload-balancer mode (the role of the reference's OVERLOAD_BALANCER
 * bracketing LP in the default refiner chain, presets.cc refinement
 * algorithms list): same deterministic schedule and commit as
 * kmp_lp_refine, but a vertex whose block exceeds its cap loses "stay" as
 * a candidate, so overloaded blocks shed boundary vertices to the best
 * admissible targets even at negative gain. Use before kmp_lp_refine when
 * the input partition may violate the caps. Returns the edge cut, or -1. */
/* Underload-balancer mode (presets.cc:332-338 UNDERLOAD_BALANCER;
 * refinement/balancer/underload_balancer.cc semantics): fill blocks below
 * min_block_weights[b] with best-gain admissible vertices, never dropping a
 * source below its own minimum and never overshooting any maximum. A no-op
 * when all minima are already satisfied (the reference's refine() gate). */
/* Clusterer::set_communities (coarsening/clusterer.h:35,
 * lp_clusterer.cc:61-66,193-194): when set (len n), kmp_lp_cluster never
 * merges vertices across community boundaries. NULL clears. */
int kmp_lp_set_communities(kmp_lp_t *e, const uint32_t *communities);

/* On-GPU degree-bucket rearrangement (permutator.cc:36-110 semantics,
 * bit-identical to kmp_rearrange_degree_buckets): rebuilds the engine's
 * device CSR in log2-degree-bucket order, writes perm_out[u_old] = u_new
 * (len n). Call before refine/cluster; map partitions through perm_out. */
int kmp_lp_rearrange_degree_buckets(kmp_lp_t *e, uint32_t *perm_out);

int64_t kmp_lp_underload(
    kmp_lp_t *e,
    uint32_t k,
    const int64_t *max_block_weights,
    const int64_t *min_block_weights,
    uint32_t *partition,
    uint64_t seed,
    int iters,
    kmp_lp_stats_t *stats
);

int64_t kmp_lp_balance(
    kmp_lp_t *e,
    uint32_t k,
    const int64_t *max_block_weights,
    uint32_t *partition,
    uint64_t seed,
    int iters,
    kmp_lp_stats_t *stats
);

/* Deterministic LP clustering (coarsening instantiation; clusters start as
 * singletons, uniform cap, isolated-node + two-hop passes). clustering: out,
 * n entries. Returns the number of non-empty clusters, or -1 on error. */
int64_t kmp_lp_cluster(
    kmp_lp_t *e,
    int64_t max_cluster_weight,
    uint32_t desired_clusters,
    uint32_t *clustering,
    uint64_t seed,
    int iters,
    kmp_lp_stats_t *stats
);

/* --------------------------- sharded (multi-GPU) refinement sub-steps ----
 * One process per GPU; rank r of world R computes phase A for its slice of
 * each chunk's position range, the proposal lists are all-gathered by the
 * caller (RCCL via torch.distributed), and every rank runs the identical
 * deterministic commit, so all replicas stay bit-identical (mirrors the
 * ghost-label exchange of kaminpar-dist/coarsening/clustering/lp/
 * global_lp_clusterer.cc:480-594 with labels replicated instead of owned).
 *
 * Proposals are 16-byte records {u, to, rank_in_chunk, weight} (uint32x4). */

/* Initialize a sharded refinement run. partition: host, n entries. */
int kmp_lp_refine_begin(
    kmp_lp_t *e,
    uint32_t k,
    const int64_t *max_block_weights,
    const uint32_t *partition,
    uint64_t seed
);

/* Number of chunks per sweep (fixed by the schedule). */
uint32_t kmp_lp_num_chunks(const kmp_lp_t *e);

/* Phase A for chunk `chunk` of sweep `iter`, positions [pos_lo, pos_hi).
 * Writes proposals to the DEVICE buffer d_out (capacity cap records) and
 * returns the record count (host), or -1 on error/overflow. */
int64_t kmp_lp_phase_a(
    kmp_lp_t *e, int iter, uint32_t chunk, uint32_t pos_lo, uint32_t pos_hi,
    void *d_out, uint32_t cap
);

/* Commit `count` proposals (DEVICE buffer d_props, 16B records; may be the
 * concatenation of all ranks' phase-A outputs in rank order) for chunk
 * `chunk` of sweep `iter`. Returns committed moves, or -1 on error. */
int64_t kmp_lp_commit(
    kmp_lp_t *e, int iter, uint32_t chunk, const void *d_props, uint32_t count
);

/* Finish a sharded run: download the partition, report stats. */
int64_t kmp_lp_refine_end(kmp_lp_t *e, uint32_t *partition, kmp_lp_stats_t *stats);

/* ------------------- device-resident stepping (benchmark region) ---------
 * The timed region excludes host transfers and the final edge-cut kernel,
 * matching the reference's LP-only timing
 * (shm_label_propagation_benchmark.cc:121-123). */
/* Sharded deterministic commit (multi-GPU): rank owns targets [c_lo,c_hi)
 * of the ALL-GATHERED proposal list (global rank order). Protocol per chunk
 * (mirrors kaminpar-dist/refinement/lp/lp_refiner.cc:296-333):
 *   shard_begin  -> local sort + full-admission departure contributions
 *                   into d_dep_out (k+1 i64; caller allreduce-sums = global)
 *   loop: shard_round(dep_global) -> d_delta_out (k+1; [k]=changed);
 *         caller allreduces, subtracts delta from dep_global, repeats
 *         until the summed changed flag is 0
 *   shard_finish_meta -> own targets' rank cutoffs + admitted arrivals
 *                   (caller allreduce-sums both = allgather)
 *   shard_apply  -> every rank applies the identical admitted set + weight
 *                   updates + active-set maintenance; returns moves.
 * All d_* pointers are DEVICE pointers. Bit-identical to kmp_lp_commit. */
/* Adopt an external HIP stream (e.g. torch's current stream) so the
 * multi-GPU collectives and the engine's kernels share one stream; pass
 * NULL to restore the engine's own stream. Engine must be idle. */
int kmp_lp_set_stream(kmp_lp_t *e, void *external_stream);

int kmp_lp_shard_begin(kmp_lp_t *e, uint32_t c_lo, uint32_t c_hi,
                       const void *d_props, uint32_t count, long long *d_dep_out);
int kmp_lp_shard_round(kmp_lp_t *e, uint32_t c_lo, uint32_t c_hi,
                       const long long *d_dep_global, long long *d_delta_out);
int kmp_lp_shard_finish_meta(kmp_lp_t *e, uint32_t c_lo, uint32_t c_hi,
                             unsigned long long *d_cutoff_out,
                             long long *d_arr_out);
int64_t kmp_lp_shard_apply(kmp_lp_t *e, int iter, uint32_t chunk,
                           const void *d_props, uint32_t count,
                           const unsigned long long *d_cutoff_all,
                           const long long *d_arr_all,
                           const long long *d_dep_global);

/* C++ RCCL distributed driver: the whole sharded-commit chunk loop with
 * collectives issued directly on the engine stream (the python/torch
 * orchestration has a measured ~0.65 ms/chunk host floor; this driver's
 * per-chunk host work is the fixpoint convergence readback only).
 * Bootstrap: rank 0 calls kmp_nccl_unique_id, the 128-byte id is
 * exchanged out of band (e.g. a torch.distributed broadcast), every rank
 * calls kmp_nccl_comm_init. nccl_comm may be NULL at world 1. */
int kmp_nccl_unique_id(void *out128);
void *kmp_nccl_comm_init(int world, int rank, const void *id128);
void kmp_nccl_comm_destroy(void *comm);
int64_t kmp_lp_refine_dist(
    kmp_lp_t *e,
    uint32_t k,
    const int64_t *max_block_weights,
    uint32_t *partition,
    uint64_t seed,
    int iters,
    void *nccl_comm,
    int rank,
    int world,
    kmp_lp_stats_t *stats
);

int kmp_lp_reset(kmp_lp_t *e);                     /* restore initial state (D2D) */
int64_t kmp_lp_run_sweeps(kmp_lp_t *e, int iters); /* the timed LP region */
int kmp_lp_get_stats(kmp_lp_t *e, kmp_lp_stats_t *stats); /* no cut/download */

#ifdef __cplusplus
} /* extern "C" */
#endif

#endif /* KAMINPAR_LP_H */
/* ---------------------------------------------------------------- v1.1 ---
 * Cluster contraction on the GPU (restates
 * kaminpar-shm/coarsening/contraction/ semantics: coarse id = prefix rank of
 * occupied cluster ids, intra-cluster edges dropped, parallel edges merged
 * with summed weights, coarse adjacency sorted by (cu, cv)). Returns the
 * coarse node count; *coarse_out receives a new host graph handle and
 * mapping_out (n entries) the fine->coarse projection. -1 on error. */
#ifdef __cplusplus
extern "C" {
#endif
int64_t kmp_contract(
    kmp_lp_t *e,
    const uint32_t *clustering,
    uint32_t *mapping_out,
    kmp_graph_t **coarse_out
);

/* Contract and hand the coarse graph DIRECTLY to a new engine without the
 * host round-trip (device-resident multilevel chain; the coarse CSR stays
 * in HBM). mapping_out (n entries, host) is still produced. The new engine
 * owns the coarse graph; use kmp_lp_download_graph when a host copy is
 * needed (e.g. the coarsest level for initial partitioning). Identical
 * results to kmp_contract + kmp_lp_create on the downloaded graph. */
int64_t kmp_contract_engine(
    kmp_lp_t *e,
    const uint32_t *clustering,
    uint32_t *mapping_out,
    kmp_lp_t **coarse_eng_out
);

/* Download an engine's device-resident CSR into a new host graph handle. */
kmp_graph_t *kmp_lp_download_graph(const kmp_lp_t *e);

uint32_t kmp_lp_n(const kmp_lp_t *e);
uint64_t kmp_lp_m(const kmp_lp_t *e);

/* ------------------------------------------------- multilevel pipeline */

/* Recursive-bisection initial partitioning on a (small) host graph: greedy
 * graph growing from `reps` distinct high-degree seeds per bisection, each
 * polished by two-way FM with best-prefix rollback, then a gain-aware
 * overload balancer. Restates the reference's initial-partitioning recipe
 * in simplified form (kaminpar-shm/initial_partitioning/). CPU-only. */
int kmp_initial_partition(
    const kmp_graph_t *g,
    uint32_t k,
    int64_t max_block_weight,
    int reps,
    uint32_t *part_out
);

/* Full multilevel partition on the GPU (BASELINE config 3; the shape of
 * KaMinPar::compute_partition, kaminpar.h:970-1025): GPU LP clustering +
 * device-resident contraction per level, CPU initial partitioning on the
 * coarsest graph, GPU LP refinement at every level. Deterministic;
 * bit-identical to the Python driver kaminpar_amd.partition.partition.
 * Returns the final edge cut, or -1 on error. */
int64_t kmp_partition(
    const kmp_graph_t *g,
    uint32_t k,
    double eps,
    uint64_t seed,
    int iters,
    uint32_t contraction_limit, /* 0 -> default 2000 */
    uint32_t stop_n,            /* 0 -> default 512 */
    int ip_reps,                /* 0 -> default 8 */
    uint32_t *part_out
);

/* Progressive-k extension helper (C twin of partition.py
 * _extend_partition): split block groups in half via FM-polished subset
 * bisections while every block keeps >= split_c vertices (or force != 0).
 * group_lo/group_w are k-sized arrays holding *num_groups entries. */
int kmp_extend_partition(
    const kmp_graph_t *g, uint32_t *part, uint32_t k, int64_t mbw_val,
    uint32_t split_c, int reps, int force, uint32_t *group_lo,
    uint32_t *group_w, uint32_t *num_groups
);

/* Overload balancer (uniform cap) on a host graph, in place. */
int kmp_balance_partition(
    const kmp_graph_t *g, uint32_t k, int64_t cap, uint32_t *part
);

/* Deterministic k-way boundary FM with pass-level best-prefix rollback
 * (serial-deterministic restatement of the reference's k-way FM refiner
 * shape, kaminpar-shm/refinement/fm/fm_refiner.cc). caps[k] are hard
 * per-block weight caps (0 closes a block). 0 -> defaults: max_passes 3,
 * max_fruitless 300. CPU-only; the multilevel drivers run it per level on
 * graphs <= ~2M fine vertices. */
int kmp_kway_fm(
    const kmp_graph_t *g, uint32_t k, const int64_t *caps, uint32_t *part,
    int max_passes, int max_fruitless
);

/* Flat / multilevel FM bisection of a vertex subset (see partition_host.cpp). */
int kmp_bisect_subset(
    const kmp_graph_t *g, const uint32_t *nodes, uint32_t n_sub,
    int64_t target1, int64_t cap1, int64_t cap2, int reps, uint8_t *side_out
);
int kmp_bisect_subset_ml(
    const kmp_graph_t *g, const uint32_t *nodes, uint32_t n_sub,
    int64_t target1, int64_t cap1, int64_t cap2, int reps, uint8_t *side_out
);

/* Progressive-k (deep) multilevel partition: grows k by block bisections
 * during uncoarsening (the shape of the reference's deep multilevel mode,
 * kaminpar-shm/partitioning/deep/deep_multilevel.cc). Bit-identical to the
 * Python driver kaminpar_amd.partition.partition_deep. Better cuts than
 * kmp_partition on every golden case. 0 defaults:
 * contraction_limit 2000, stop_n 512, split_c auto (262144 up to 2M
 * vertices, 2000 beyond), ip_reps 8. split_c >= n selects the full
 * late-split quality mode on heavy-tailed graphs: every split at the
 * finest level (measured at R-MAT scale 23: cut 0.44x the reference's
 * best seed, at minutes of host-side bisection cost -- see DESIGN.md). */
int64_t kmp_partition_deep(
    const kmp_graph_t *g,
    uint32_t k,
    double eps,
    uint64_t seed,
    int iters,
    uint32_t contraction_limit,
    uint32_t stop_n,
    uint32_t split_c,
    int ip_reps,
    uint32_t *part_out
);

/* --------------------------------------------- ckaminpar-shaped C shim
 * Mirrors the reference's public C interface (include/kaminpar-shm/
 * ckaminpar.h:61-132: kaminpar_create / kaminpar_copy_graph /
 * kaminpar_set_k / kaminpar_set_uniform_max_block_weights /
 * kaminpar_compute_partition / kaminpar_free) with the same call order,
 * argument meaning and default types (NodeID/EdgeID u32, weights i32).
 * num_threads is accepted for signature parity; the implementation runs on
 * the GPU. */
typedef struct kaminpar_amd_t kaminpar_amd_t;

kaminpar_amd_t *kaminpar_amd_create(int num_threads);
void kaminpar_amd_free(kaminpar_amd_t *shm);
void kaminpar_amd_reseed(kaminpar_amd_t *shm, int seed);
void kaminpar_amd_copy_graph(
    kaminpar_amd_t *shm,
    uint32_t n,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,   /* NULL for unit weights */
    const int32_t *adjwgt  /* NULL for unit weights */
);
void kaminpar_amd_set_k(kaminpar_amd_t *shm, uint32_t k);
void kaminpar_amd_set_uniform_max_block_weights(kaminpar_amd_t *shm, double epsilon);
/* Returns the final edge cut, or -1 on error (missing GPU, no graph). */
void kaminpar_amd_set_absolute_max_block_weights(
    kaminpar_amd_t *shm, const int64_t *weights, uint32_t count); /* kaminpar.h:961 */
void kaminpar_amd_set_uniform_min_block_weights(
    kaminpar_amd_t *shm, double min_epsilon); /* kaminpar.h:965; context.cc:72-80 */
void kaminpar_amd_set_absolute_min_block_weights(
    kaminpar_amd_t *shm, const int64_t *weights, uint32_t count); /* kaminpar.h:966 */
void kaminpar_amd_clear_min_block_weights(kaminpar_amd_t *shm); /* kaminpar.h:968 */

int64_t kaminpar_amd_compute_partition(kaminpar_amd_t *shm, uint32_t *partition);

#ifdef __cplusplus
}
#endif
