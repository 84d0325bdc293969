// C wrapper around the FULL reference shm partitioner (kaminpar::KaMinPar,
// include/kaminpar-shm/kaminpar.h:857-1050), compiled with serial TBB stubs:
// the reference's deterministic 1-thread compute_partition, used to generate
// golden full-pipeline cuts for the config-3 multilevel comparison.
#include <cstdint>
#include <memory>
#include <vector>

#include "kaminpar-shm/kaminpar.h"

using namespace kaminpar;
using namespace kaminpar::shm;

extern "C" {

// Full multilevel partition (default preset). partition: out, n entries.
// Returns the edge cut reported by compute_partition.
int64_t kref_compute_partition(
    const uint32_t n,
    const uint64_t m,
    const uint32_t *xadj,
    const uint32_t *adjncy,
    const int32_t *vwgt,
    const int32_t *adjwgt,
    const uint32_t k,
    const double epsilon,
    const int seed,
    uint32_t *partition
) {
  KaMinPar::reseed(seed);
  KaMinPar shm(1, create_default_context());
  shm.set_output_level(OutputLevel::QUIET);

  std::vector<EdgeID> x(n + 1);
  for (uint32_t i = 0; i <= n; ++i) {
    x[i] = xadj[i];
  }
  std::vector<NodeID> a(m);
  for (uint64_t e = 0; e < m; ++e) {
    a[e] = adjncy[e];
  }
  std::vector<NodeWeight> vw;
  std::vector<EdgeWeight> ew;
  if (vwgt) {
    vw.assign(vwgt, vwgt + n);
  }
  if (adjwgt) {
    ew.assign(adjwgt, adjwgt + m);
  }

  shm.copy_graph(
      std::span<const EdgeID>(x.data(), n + 1), std::span<const NodeID>(a.data(), m),
      vwgt ? std::span<const NodeWeight>(vw.data(), n) : std::span<const NodeWeight>(),
      adjwgt ? std::span<const EdgeWeight>(ew.data(), m) : std::span<const EdgeWeight>()
  );
  shm.set_k(k);
  shm.set_uniform_max_block_weights(epsilon);

  std::vector<BlockID> part(n);
  const EdgeWeight cut = shm.compute_partition(std::span<BlockID>(part.data(), n));
  for (uint32_t u = 0; u < n; ++u) {
    partition[u] = part[u];
  }
  return static_cast<int64_t>(cut);
}

} // extern "C"
