/* Plain-C caller of the drop-in boundary (include/kaminpar_lp.h): exercises
 * the ckaminpar-shaped shim (ckaminpar.h:61-132 call order) including the
 * round-2 additions -- per-block max weights (kaminpar.h:961), min block
 * weights (kaminpar.h:965-968, chained into the underload balancer), and
 * Clusterer::set_communities (clusterer.h:35) -- compiled with gcc as C11
 * and linked against libkaminpar_lp.so. Run by
 * tests/test_gpu_parity.py::test_c_api_caller on the GPU box.
 *
 * Exit code 0 = all checks passed; prints one line per check. */
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "../include/kaminpar_lp.h"

#define CHECK(cond, what)                                                      \
  do {                                                                         \
    if (!(cond)) {                                                             \
      fprintf(stderr, "FAIL: %s\n", what);                                     \
      return 1;                                                                \
    }                                                                          \
    printf("ok: %s\n", what);                                                  \
  } while (0)

/* small ring-of-cliques graph: k cliques of size c, ring-linked */
static void build_graph(uint32_t cliques, uint32_t csz, uint32_t **xadj_out,
                        uint32_t **adj_out, uint32_t *n_out, uint64_t *m_out) {
  const uint32_t n = cliques * csz;
  uint32_t *deg = calloc(n + 1, sizeof(uint32_t));
  /* arcs: clique-internal (csz-1 each) + 2 ring arcs per clique rep */
  uint64_t m = (uint64_t)n * (csz - 1) + 2ull * cliques;
  uint32_t *xadj = malloc((n + 1) * sizeof(uint32_t));
  uint32_t *adj = malloc(m * sizeof(uint32_t));
  for (uint32_t u = 0; u < n; ++u) {
    deg[u] = csz - 1 + (u % csz == 0 ? 2 : 0);
  }
  xadj[0] = 0;
  for (uint32_t u = 0; u < n; ++u) {
    xadj[u + 1] = xadj[u] + deg[u];
  }
  for (uint32_t q = 0; q < cliques; ++q) {
    for (uint32_t i = 0; i < csz; ++i) {
      const uint32_t u = q * csz + i;
      uint32_t w = xadj[u];
      for (uint32_t j = 0; j < csz; ++j) {
        if (j != i) {
          adj[w++] = q * csz + j;
        }
      }
      if (i == 0) {
        adj[w++] = ((q + 1) % cliques) * csz;
        adj[w++] = ((q + cliques - 1) % cliques) * csz;
      }
    }
  }
  free(deg);
  *xadj_out = xadj;
  *adj_out = adj;
  *n_out = n;
  *m_out = m;
}

int main(void) {
  uint32_t *xadj, *adj, n;
  uint64_t m;
  const uint32_t cliques = 64, csz = 16, k = 8;
  build_graph(cliques, csz, &xadj, &adj, &n, &m);

  /* ---- ckaminpar-shaped shim, reference call order ---- */
  kaminpar_amd_t *shm = kaminpar_amd_create(1);
  CHECK(shm != NULL, "kaminpar_amd_create");
  kaminpar_amd_reseed(shm, 1);
  kaminpar_amd_copy_graph(shm, n, xadj, adj, NULL, NULL);
  kaminpar_amd_set_k(shm, k);
  kaminpar_amd_set_uniform_max_block_weights(shm, 0.03);

  uint32_t *part = malloc(n * sizeof(uint32_t));
  int64_t cut = kaminpar_amd_compute_partition(shm, part);
  CHECK(cut >= 0, "compute_partition (uniform caps)");
  /* k cliques of 16 into 8 blocks: the ring cut can be as low as 8 */
  CHECK(cut <= 4 * (int64_t)cliques, "cut sane for ring-of-cliques");

  /* partition ids in range */
  for (uint32_t u = 0; u < n; ++u) {
    if (part[u] >= k) {
      fprintf(stderr, "FAIL: partition id out of range\n");
      return 1;
    }
  }
  printf("ok: partition ids in range\n");

  /* ---- per-block max weights (kaminpar.h:961) ---- */
  {
    int64_t caps[8];
    for (uint32_t b = 0; b < k; ++b) {
      caps[b] = (b == 0) ? (int64_t)(csz * 4) : (int64_t)(csz * 16);
    }
    kaminpar_amd_set_absolute_max_block_weights(shm, caps, k);
    cut = kaminpar_amd_compute_partition(shm, part);
    CHECK(cut >= 0, "compute_partition (absolute per-block caps)");
    int64_t bw[8] = {0};
    for (uint32_t u = 0; u < n; ++u) {
      bw[part[u]] += 1;
    }
    for (uint32_t b = 0; b < k; ++b) {
      if (bw[b] > caps[b]) {
        fprintf(stderr, "FAIL: per-block cap violated (b=%u)\n", b);
        return 1;
      }
    }
    printf("ok: per-block caps respected\n");
  }

  /* ---- min block weights -> underload balancer (kaminpar.h:965) ---- */
  {
    kaminpar_amd_set_uniform_max_block_weights(shm, 0.20);
    kaminpar_amd_set_uniform_min_block_weights(shm, 0.50);
    cut = kaminpar_amd_compute_partition(shm, part);
    CHECK(cut >= 0, "compute_partition (min weights set)");
    int64_t bw[8] = {0};
    for (uint32_t u = 0; u < n; ++u) {
      bw[part[u]] += 1;
    }
    const int64_t minw = (int64_t)((1.0 - 0.50) * n / k + 0.999);
    for (uint32_t b = 0; b < k; ++b) {
      if (bw[b] < minw) {
        fprintf(stderr, "FAIL: per-block minimum violated (b=%u)\n", b);
        return 1;
      }
    }
    printf("ok: per-block minimums respected\n");
    kaminpar_amd_clear_min_block_weights(shm);
  }

  /* ---- Clusterer seam: set_communities (clusterer.h:35) ---- */
  {
    kmp_graph_t *g = kmp_graph_from_csr(n, m, xadj, adj, NULL, NULL);
    CHECK(g != NULL, "kmp_graph_from_csr");
    kmp_lp_t *e = kmp_lp_create(g);
    CHECK(e != NULL, "kmp_lp_create");
    uint32_t *comm = malloc(n * sizeof(uint32_t));
    for (uint32_t u = 0; u < n; ++u) {
      comm[u] = (u / csz) & 1; /* alternating clique communities */
    }
    CHECK(kmp_lp_set_communities(e, comm) == 0, "kmp_lp_set_communities");
    uint32_t *clus = malloc(n * sizeof(uint32_t));
    int64_t nc = kmp_lp_cluster(e, csz * 4, 0, clus, 1, 5, NULL);
    CHECK(nc > 0, "kmp_lp_cluster with communities");
    for (uint32_t u = 0; u < n; ++u) {
      if (comm[clus[u]] != comm[u]) {
        fprintf(stderr, "FAIL: cluster crosses a community (u=%u)\n", u);
        return 1;
      }
    }
    printf("ok: no cluster crosses a community\n");
    kmp_lp_set_communities(e, NULL);
    kmp_lp_free(e);
    kmp_graph_free(g);
    free(comm);
    free(clus);
  }

  kaminpar_amd_free(shm);
  free(xadj);
  free(adj);
  free(part);
  printf("c_api_check: ALL OK\n");
  return 0;
}
