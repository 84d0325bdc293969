// C wrapper around the REFERENCE KaMinPar label-propagation hot path,
// compiled from the sources under /root/reference with serial TBB stubs
// (oracle/_ref_build/stubs). Single-threaded and deterministic: this is the
// reference's own 1-thread behaviour, which the reference pins in
// tests/endtoend/shm_endtoend_test.cc:189-217 (determinism under a seed).
//
// Exposes:
//   kref_lp_cluster  - LPClustering::compute_clustering
//                      (kaminpar-shm/coarsening/clustering/lp_clusterer.cc:395)
//   kref_lp_refine   - LabelPropagationRefiner::refine
//                      (kaminpar-shm/refinement/lp/lp_refiner.cc:370)
//   kref_edge_cut    - metrics::edge_cut_seq (kaminpar-shm/metrics.cc:73)
//
// Used ONLY as a parity oracle (oracle/_ref/libkaminpar_ref.so); never on the
// product path.

#include <cstdint>
#include <cstring>
#include <memory>
#include <vector>

#include "kaminpar-shm/coarsening/clustering/lp_clusterer.h"
#include "kaminpar-shm/datastructures/csr_graph.h"
#include "kaminpar-shm/datastructures/graph.h"
#include "kaminpar-shm/datastructures/partitioned_graph.h"
#include "kaminpar-shm/kaminpar.h"
#include "kaminpar-shm/metrics.h"
#include "kaminpar-shm/refinement/lp/lp_refiner.h"

#include "kaminpar-common/datastructures/static_array.h"
#include "kaminpar-common/random.h"

using namespace kaminpar;
using namespace kaminpar::shm;

namespace {

CSRGraph make_csr_graph(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt
) {
  StaticArray<EdgeID> nodes(n + 1);
  for (uint32_t i = 0; i <= n; ++i) {
    nodes[i] = xadj[i];
  }
  StaticArray<NodeID> edges(m);
  for (uint64_t e = 0; e < m; ++e) {
    edges[e] = adjncy[e];
  }
  StaticArray<NodeWeight> node_weights;
  if (vwgt != nullptr) {
    node_weights.resize(n);
    for (uint32_t i = 0; i < n; ++i) {
      node_weights[i] = vwgt[i];
    }
  }
  StaticArray<EdgeWeight> edge_weights;
  if (adjwgt != nullptr) {
    edge_weights.resize(m);
    for (uint64_t e = 0; e < m; ++e) {
      edge_weights[e] = adjwgt[e];
    }
  }
  return CSRGraph(
      std::move(nodes), std::move(edges), std::move(node_weights), std::move(edge_weights)
  );
}

} // namespace

extern "C" {

// LP clustering (coarsening instantiation). Returns 0 on success.
// out_clustering must have n entries.
int kref_lp_cluster(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt,
    const int seed,
    const int num_iterations,
    const int64_t max_cluster_weight,
    const uint32_t desired_num_clusters,
    uint32_t *out_clustering
) {
  Random::reseed(seed);

  Context ctx = create_default_context();
  if (num_iterations > 0) {
    ctx.coarsening.clustering.lp.num_iterations = num_iterations;
  }

  Graph graph(std::make_unique<CSRGraph>(make_csr_graph(n, m, xadj, adjncy, vwgt, adjwgt)));

  LPClustering clusterer(ctx.coarsening);
  clusterer.set_max_cluster_weight(static_cast<NodeWeight>(max_cluster_weight));
  clusterer.set_desired_cluster_count(desired_num_clusters);

  StaticArray<NodeID> clustering(n);
  clusterer.compute_clustering(clustering, graph, false);

  for (uint32_t u = 0; u < n; ++u) {
    out_clustering[u] = clustering[u];
  }
  return 0;
}

// LP refinement (k-way instantiation). partition is in/out (n entries).
// Returns the resulting edge cut (sequentially recomputed), or -1 on error.
int64_t kref_lp_refine(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt,
    const uint32_t k,
    const double epsilon,
    const int seed,
    const int num_iterations,
    uint32_t *partition
) {
  Random::reseed(seed);

  Context ctx = create_default_context();
  if (num_iterations > 0) {
    ctx.refinement.lp.num_iterations = num_iterations;
  }

  Graph graph(std::make_unique<CSRGraph>(make_csr_graph(n, m, xadj, adjncy, vwgt, adjwgt)));
  ctx.partition.setup(graph, static_cast<BlockID>(k), epsilon);

  StaticArray<BlockID> part(n);
  for (uint32_t u = 0; u < n; ++u) {
    part[u] = partition[u];
  }
  PartitionedGraph p_graph(graph, static_cast<BlockID>(k), std::move(part));

  LabelPropagationRefiner refiner(ctx);
  refiner.initialize(p_graph);
  refiner.refine(p_graph, ctx.partition);

  for (uint32_t u = 0; u < n; ++u) {
    partition[u] = p_graph.block(u);
  }
  return static_cast<int64_t>(metrics::edge_cut(p_graph));
}

// Max block weight as the reference's PartitionContext computes it
// (kaminpar-shm/context.cc:27-39).
int64_t kref_max_block_weight(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt,
    const uint32_t k,
    const double epsilon
) {
  Graph graph(std::make_unique<CSRGraph>(make_csr_graph(n, m, xadj, adjncy, vwgt, adjwgt)));
  PartitionContext p_ctx;
  p_ctx.setup(graph, static_cast<BlockID>(k), epsilon);
  return p_ctx.max_block_weight(0);
}

// Sequential edge cut of a partition (kaminpar-shm/metrics.cc:73-84).
int64_t kref_edge_cut(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *adjwgt,
    const uint32_t *partition
) {
  int64_t cut = 0;
  for (uint32_t u = 0; u < n; ++u) {
    for (uint32_t e = xadj[u]; e < xadj[u + 1]; ++e) {
      if (partition[u] != partition[adjncy[e]]) {
        cut += (adjwgt != nullptr) ? adjwgt[e] : 1;
      }
    }
  }
  return cut / 2;
}

} // extern "C"

#include "kaminpar-shm/coarsening/contraction/cluster_contraction.h"

extern "C" {

// Contract a clustering with the reference implementation; returns the
// coarse node count and fills mapping + the coarse CSR with each coarse
// adjacency list SORTED by target (the reference's own in-list order is
// scheduling-dependent; sorting canonicalizes for comparison).
int64_t kref_contract(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt,
    const uint32_t *clustering,
    uint32_t *mapping_out,
    uint32_t *c_xadj,
    uint32_t *c_adjncy,
    int32_t *c_vwgt,
    int32_t *c_adjwgt,
    uint64_t *c_m_out
) {
  Graph graph(std::make_unique<CSRGraph>(make_csr_graph(n, m, xadj, adjncy, vwgt, adjwgt)));

  StaticArray<NodeID> clus(n);
  for (uint32_t u = 0; u < n; ++u) {
    clus[u] = clustering[u];
  }

  Context ctx = create_default_context();
  auto coarse = contract_clustering(graph, std::move(clus), ctx.coarsening.contraction);

  const auto &cg = coarse->get();
  const CSRGraph &csr = *dynamic_cast<const CSRGraph *>(cg.underlying_graph());
  const uint32_t c_n = csr.n();
  const uint64_t c_m = csr.m();

  // mapping: project a fine identity labelling upward is not exposed;
  // recover it by projecting coarse ids down (project_up maps coarse->fine)
  {
    std::vector<BlockID> coarse_ids(c_n);
    for (uint32_t c = 0; c < c_n; ++c) {
      coarse_ids[c] = c;
    }
    std::vector<BlockID> fine_ids(n);
    coarse->project_up(
        std::span<const BlockID>(coarse_ids.data(), c_n),
        std::span<BlockID>(fine_ids.data(), n)
    );
    for (uint32_t u = 0; u < n; ++u) {
      mapping_out[u] = fine_ids[u];
    }
  }

  for (uint32_t c = 0; c <= c_n; ++c) {
    c_xadj[c] = csr.raw_nodes()[c];
  }
  for (uint32_t c = 0; c < c_n; ++c) {
    c_vwgt[c] = csr.node_weight(c);
  }
  // sort each adjacency list by target for canonical comparison
  for (uint32_t c = 0; c < c_n; ++c) {
    std::vector<std::pair<uint32_t, int32_t>> row;
    for (uint64_t e = csr.raw_nodes()[c]; e < csr.raw_nodes()[c + 1]; ++e) {
      row.emplace_back(csr.raw_edges()[e], csr.edge_weight(e));
    }
    std::sort(row.begin(), row.end());
    uint64_t e = csr.raw_nodes()[c];
    for (const auto &[v, w] : row) {
      c_adjncy[e] = v;
      c_adjwgt[e] = w;
      ++e;
    }
  }
  *c_m_out = c_m;
  return static_cast<int64_t>(c_n);
}

} // extern "C"
