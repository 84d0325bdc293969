// Host-side graph handling for the MI355X LP path: CSR container, R-MAT and
// RGG2D generators (deterministic, parallel with OpenMP), METIS ASCII reader.
//
// The CSR layout is the reference's CSRGraphMemory
// (kaminpar-shm/datastructures/csr_graph.h:27-33): xadj[n+1] (u32 here),
// adjncy[m] with both arc directions stored, optional vwgt/adjwgt.
// Generators follow BASELINE.md: Graph500 R-MAT parameters
// (a=0.57, b=0.19, c=0.19, d=0.05), symmetrized, deduplicated, self-loops
// removed, unit weights, fixed seed recorded by the caller.

#include "../../include/kaminpar_lp.h"

#include <algorithm>
#include <cctype>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <type_traits>
#include <vector>

#ifdef _OPENMP
#include <omp.h>
#endif

#include "lp_common.h"

using kmp::i32;
using kmp::i64;
using kmp::splitmix64;
using kmp::u32;
using kmp::u64;

struct kmp_graph_t {
  u32 n = 0;
  u64 m = 0;
  std::vector<u32> xadj;   // empty when the graph uses 64-bit offsets
  std::vector<u64> xadj64; // EdgeID-64 offsets (m >= 2^32); else empty
  std::vector<u32> adjncy;
  std::vector<i32> vwgt;   // empty -> unit
  std::vector<i32> adjwgt; // empty -> unit
  i64 total_node_weight = 0;
};

namespace {

// Parallel LSD radix sort of u64 keys (4 passes x 16 bits).
void radix_sort_u64(std::vector<u64> &keys) {
  const size_t n = keys.size();
  if (n < (1u << 16)) {
    std::sort(keys.begin(), keys.end());
    return;
  }
  std::vector<u64> tmp(n);
  u64 *src = keys.data();
  u64 *dst = tmp.data();

  int nt = 1;
#ifdef _OPENMP
  nt = omp_get_max_threads();
#endif
  const size_t block = (n + nt - 1) / nt;
  std::vector<size_t> hist(static_cast<size_t>(nt) * 65536);

  for (int pass = 0; pass < 4; ++pass) {
    const int shift = pass * 16;
    std::fill(hist.begin(), hist.end(), 0);
#ifdef _OPENMP
#pragma omp parallel num_threads(nt)
#endif
    {
      int t = 0;
#ifdef _OPENMP
      t = omp_get_thread_num();
#endif
      size_t lo = t * block, hi = std::min(n, lo + block);
      size_t *h = &hist[static_cast<size_t>(t) * 65536];
      for (size_t i = lo; i < hi; ++i) {
        ++h[(src[i] >> shift) & 0xFFFF];
      }
    }
    // exclusive prefix over (bucket, thread)
    size_t sum = 0;
    for (int b = 0; b < 65536; ++b) {
      for (int t = 0; t < nt; ++t) {
        size_t &h = hist[static_cast<size_t>(t) * 65536 + b];
        size_t c = h;
        h = sum;
        sum += c;
      }
    }
#ifdef _OPENMP
#pragma omp parallel num_threads(nt)
#endif
    {
      int t = 0;
#ifdef _OPENMP
      t = omp_get_thread_num();
#endif
      size_t lo = t * block, hi = std::min(n, lo + block);
      size_t *h = &hist[static_cast<size_t>(t) * 65536];
      for (size_t i = lo; i < hi; ++i) {
        dst[h[(src[i] >> shift) & 0xFFFF]++] = src[i];
      }
    }
    std::swap(src, dst);
  }
  // 4 passes (even count): result is back in keys.data()
}

// Build a graph from a deduplicated, sorted arc list (both directions
// present, no self loops).
kmp_graph_t *graph_from_sorted_arcs(u32 n, const std::vector<u64> &arcs) {
  auto *g = new kmp_graph_t();
  g->n = n;
  g->m = arcs.size();
  g->xadj.assign(static_cast<size_t>(n) + 1, 0);
  g->adjncy.resize(arcs.size());
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long long i = 0; i < static_cast<long long>(arcs.size()); ++i) {
    g->adjncy[i] = static_cast<u32>(arcs[i] & 0xFFFFFFFFu);
  }
  for (u64 a : arcs) {
    ++g->xadj[(a >> 32) + 1];
  }
  for (u32 u = 0; u < n; ++u) {
    g->xadj[u + 1] += g->xadj[u];
  }
  g->total_node_weight = n;
  return g;
}

kmp_graph_t *graph_from_pairs(u32 n, std::vector<u64> &arcs) {
  radix_sort_u64(arcs);
  arcs.erase(std::unique(arcs.begin(), arcs.end()), arcs.end());
  return graph_from_sorted_arcs(n, arcs);
}

} // namespace

extern "C" {

/* EdgeID-64 construction path (ckaminpar.h:33-37 KAMINPAR_64BIT_EDGE_IDS
 * analogue): graphs with >= 2^32 directed arcs carry 64-bit offsets. */
kmp_graph_t *kmp_graph_from_csr64(
    u32 n, u64 m, const u64 *xadj, const u32 *adjncy, const i32 *vwgt, const i32 *adjwgt
) {
  if (xadj == nullptr || (m > 0 && adjncy == nullptr) || xadj[n] != m) {
    return nullptr;
  }
  auto *g = new kmp_graph_t();
  g->n = n;
  g->m = m;
  g->xadj64.assign(xadj, xadj + n + 1);
  g->adjncy.assign(adjncy, adjncy + m);
  if (vwgt) {
    g->vwgt.assign(vwgt, vwgt + n);
    g->total_node_weight = 0;
    for (u32 u = 0; u < n; ++u) {
      g->total_node_weight += vwgt[u];
    }
  } else {
    g->total_node_weight = n;
  }
  if (adjwgt) {
    g->adjwgt.assign(adjwgt, adjwgt + m);
  }
  return g;
}

kmp_graph_t *kmp_graph_from_csr(
    u32 n, u64 m, const u32 *xadj, const u32 *adjncy, const i32 *vwgt, const i32 *adjwgt
) {
  if (xadj == nullptr || (m > 0 && adjncy == nullptr) || xadj[n] != m) {
    return nullptr;
  }
  auto *g = new kmp_graph_t();
  g->n = n;
  g->m = m;
  g->xadj.assign(xadj, xadj + n + 1);
  g->adjncy.assign(adjncy, adjncy + m);
  if (vwgt) {
    g->vwgt.assign(vwgt, vwgt + n);
    g->total_node_weight = 0;
    for (u32 u = 0; u < n; ++u) {
      g->total_node_weight += vwgt[u];
    }
  } else {
    g->total_node_weight = n;
  }
  if (adjwgt) {
    g->adjwgt.assign(adjwgt, adjwgt + m);
  }
  return g;
}

kmp_graph_t *kmp_gen_rmat(int scale, int edgefactor, u64 seed) {
  const u32 n = 1u << scale;
  const u64 num_edges = static_cast<u64>(edgefactor) << scale;
  std::vector<u64> arcs(2 * num_edges);

#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long long e = 0; e < static_cast<long long>(num_edges); ++e) {
    u64 s = kmp::mix_seed(seed, 0x524D4154ULL + static_cast<u64>(e));
    u32 u = 0, v = 0;
    for (int level = 0; level < scale; ++level) {
      s = splitmix64(s);
      // Graph500 quadrant probabilities a=0.57 b=0.19 c=0.19 d=0.05,
      // thresholds on a 32-bit draw.
      const u32 r = static_cast<u32>(s >> 32);
      u32 q;
      if (r < 2448131359u) { // 0.57 * 2^32
        q = 0;
      } else if (r < 3264175145u) { // (0.57+0.19) * 2^32
        q = 1;
      } else if (r < 4080218931u) { // (0.57+0.38) * 2^32
        q = 2;
      } else {
        q = 3;
      }
      u = (u << 1) | (q >> 1);
      v = (v << 1) | (q & 1);
    }
    if (u == v) {
      // self loop: drop by emitting a sentinel arc that dedup removes
      // (u,u) -> keep as self loop marker; filtered below via same key twice
      arcs[2 * e] = ~0ULL;
      arcs[2 * e + 1] = ~0ULL;
    } else {
      arcs[2 * e] = (static_cast<u64>(u) << 32) | v;
      arcs[2 * e + 1] = (static_cast<u64>(v) << 32) | u;
    }
  }

  radix_sort_u64(arcs);
  // drop sentinel (~0) tail and duplicates
  while (!arcs.empty() && arcs.back() == ~0ULL) {
    arcs.pop_back();
  }
  arcs.erase(std::unique(arcs.begin(), arcs.end()), arcs.end());
  return graph_from_sorted_arcs(n, arcs);
}

kmp_graph_t *kmp_gen_rgg2d(u32 n, double avg_deg, u64 seed) {
  // radius for expected average degree: pi r^2 n = avg_deg
  const double r = std::sqrt(avg_deg / (M_PI * n));
  const u32 grid = std::max<u32>(1, static_cast<u32>(1.0 / r));
  const double cell = 1.0 / grid;

  std::vector<float> xs(n), ys(n);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long long i = 0; i < static_cast<long long>(n); ++i) {
    xs[i] = static_cast<float>(
        (splitmix64(kmp::mix_seed(seed, 2 * i)) >> 11) * (1.0 / 9007199254740992.0)
    );
    ys[i] = static_cast<float>(
        (splitmix64(kmp::mix_seed(seed, 2 * i + 1)) >> 11) * (1.0 / 9007199254740992.0)
    );
  }

  // cell bucketing
  std::vector<u32> cell_of(n), cell_count(static_cast<size_t>(grid) * grid + 1, 0);
  for (u32 i = 0; i < n; ++i) {
    u32 cx = std::min<u32>(grid - 1, static_cast<u32>(xs[i] / cell));
    u32 cy = std::min<u32>(grid - 1, static_cast<u32>(ys[i] / cell));
    cell_of[i] = cy * grid + cx;
    ++cell_count[cell_of[i] + 1];
  }
  for (size_t c = 0; c < static_cast<size_t>(grid) * grid; ++c) {
    cell_count[c + 1] += cell_count[c];
  }
  std::vector<u32> by_cell(n);
  {
    std::vector<u32> cur(cell_count.begin(), cell_count.end() - 1);
    for (u32 i = 0; i < n; ++i) {
      by_cell[cur[cell_of[i]]++] = i;
    }
  }

  const float r2 = static_cast<float>(r * r);
  std::vector<std::vector<u64>> locals;
  int nt = 1;
#ifdef _OPENMP
  nt = omp_get_max_threads();
#endif
  locals.resize(nt);

#ifdef _OPENMP
#pragma omp parallel num_threads(nt)
#endif
  {
    int t = 0;
#ifdef _OPENMP
    t = omp_get_thread_num();
#endif
    auto &out = locals[t];
#ifdef _OPENMP
#pragma omp for schedule(dynamic, 64)
#endif
    for (long long ci = 0; ci < static_cast<long long>(grid) * grid; ++ci) {
      const u32 cx = ci % grid, cy = ci / grid;
      for (u32 a = cell_count[ci]; a < cell_count[ci + 1]; ++a) {
        const u32 i = by_cell[a];
        for (int dy = -1; dy <= 1; ++dy) {
          for (int dx = -1; dx <= 1; ++dx) {
            const int nx = static_cast<int>(cx) + dx, ny = static_cast<int>(cy) + dy;
            if (nx < 0 || ny < 0 || nx >= static_cast<int>(grid) || ny >= static_cast<int>(grid)) {
              continue;
            }
            const size_t cj = static_cast<size_t>(ny) * grid + nx;
            for (u32 b = cell_count[cj]; b < cell_count[cj + 1]; ++b) {
              const u32 j = by_cell[b];
              if (j == i) {
                continue;
              }
              const float ddx = xs[i] - xs[j], ddy = ys[i] - ys[j];
              if (ddx * ddx + ddy * ddy < r2) {
                out.push_back((static_cast<u64>(i) << 32) | j);
              }
            }
          }
        }
      }
    }
  }

  size_t total = 0;
  for (auto &l : locals) {
    total += l.size();
  }
  std::vector<u64> arcs;
  arcs.reserve(total);
  for (auto &l : locals) {
    arcs.insert(arcs.end(), l.begin(), l.end());
    l.clear();
    l.shrink_to_fit();
  }
  return graph_from_pairs(n, arcs);
}

kmp_graph_t *kmp_read_metis(const char *path) {
  FILE *f = fopen(path, "r");
  if (!f) {
    return nullptr;
  }
  char line[1 << 16];
  u32 n = 0;
  u64 m2 = 0;
  int fmt = 0;
  // header
  while (fgets(line, sizeof(line), f)) {
    if (line[0] == '%') {
      continue;
    }
    unsigned long long nn = 0, mm = 0;
    int cnt = sscanf(line, "%llu %llu %d", &nn, &mm, &fmt);
    if (cnt < 2) {
      fclose(f);
      return nullptr;
    }
    n = static_cast<u32>(nn);
    m2 = mm;
    break;
  }
  const bool has_vwgt = (fmt == 10 || fmt == 11);
  const bool has_adjwgt = (fmt == 1 || fmt == 11);

  auto *g = new kmp_graph_t();
  g->n = n;
  g->xadj.assign(static_cast<size_t>(n) + 1, 0);
  g->adjncy.reserve(2 * m2);
  if (has_vwgt) {
    g->vwgt.resize(n);
  }
  if (has_adjwgt) {
    g->adjwgt.reserve(2 * m2);
  }

  u32 u = 0;
  while (u < n && fgets(line, sizeof(line), f)) {
    if (line[0] == '%') {
      continue;
    }
    char *p = line;
    auto next_tok = [&]() -> long long {
      while (*p && std::isspace(static_cast<unsigned char>(*p))) {
        ++p;
      }
      if (!*p) {
        return -1;
      }
      long long v = strtoll(p, &p, 10);
      return v;
    };
    if (has_vwgt) {
      long long w = next_tok();
      g->vwgt[u] = w < 0 ? 1 : static_cast<i32>(w);
    }
    while (true) {
      long long v = next_tok();
      if (v < 0) {
        break;
      }
      g->adjncy.push_back(static_cast<u32>(v - 1)); // 1-based in file
      if (has_adjwgt) {
        long long w = next_tok();
        g->adjwgt.push_back(static_cast<i32>(w));
      }
      ++g->xadj[u + 1];
    }
    ++u;
  }
  fclose(f);
  for (u32 i = 0; i < n; ++i) {
    g->xadj[i + 1] += g->xadj[i];
  }
  g->m = g->adjncy.size();
  g->total_node_weight = 0;
  if (has_vwgt) {
    for (u32 i = 0; i < n; ++i) {
      g->total_node_weight += g->vwgt[i];
    }
  } else {
    g->total_node_weight = n;
  }
  return g;
}

u32 kmp_graph_n(const kmp_graph_t *g) { return g->n; }
u64 kmp_graph_m(const kmp_graph_t *g) { return g->m; }
const u32 *kmp_graph_xadj(const kmp_graph_t *g) {
  return g->xadj64.empty() ? g->xadj.data() : nullptr;
}
const u64 *kmp_graph_xadj64(const kmp_graph_t *g) {
  return g->xadj64.empty() ? nullptr : g->xadj64.data();
}
const u32 *kmp_graph_adjncy(const kmp_graph_t *g) { return g->adjncy.data(); }
const i32 *kmp_graph_vwgt(const kmp_graph_t *g) {
  return g->vwgt.empty() ? nullptr : g->vwgt.data();
}
const i32 *kmp_graph_adjwgt(const kmp_graph_t *g) {
  return g->adjwgt.empty() ? nullptr : g->adjwgt.data();
}
i64 kmp_graph_total_node_weight(const kmp_graph_t *g) { return g->total_node_weight; }

void kmp_graph_free(kmp_graph_t *g) { delete g; }

i64 kmp_edge_cut_host(const kmp_graph_t *g, const u32 *labels) {
  i64 cut = 0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static) reduction(+ : cut)
#endif
  for (long long u = 0; u < static_cast<long long>(g->n); ++u) {
    for (u64 e = g->xadj[u]; e < g->xadj[u + 1]; ++e) {
      if (labels[u] != labels[g->adjncy[e]]) {
        cut += g->adjwgt.empty() ? 1 : g->adjwgt[e];
      }
    }
  }
  return cut / 2;
}

// Expose the product-side Feistel permutation (lp_common.h, the same code
// the GPU kernels execute) for cross-checking against the oracle's
// independent restatement.
// out must have kmp::pos_count(n) entries; values >= n mark skipped tail
// positions of the last 64-vertex unit.
void kmp_perm(u32 n, u64 seed, int iter, u32 *out) {
  const kmp::BlockPerm perm(n, kmp::iter_seed_of(seed, iter));
  const u32 P = kmp::pos_count(n);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
  for (long long p = 0; p < static_cast<long long>(P); ++p) {
    out[p] = perm(static_cast<u32>(p));
  }
}

i64 kmp_max_block_weight(const kmp_graph_t *g, u32 k, double eps) {
  // context.cc:27-39: (1+eps) * ceil(total_node_weight / k), truncated to int
  const double pbw = std::ceil(1.0 * g->total_node_weight / k);
  return static_cast<i64>((1.0 + eps) * pbw);
}

// METIS ASCII writer (kaminpar-io metis format; see the reader below and
// docs/graph_file_format.md "METIS Graph File Format").
int kmp_write_metis(const kmp_graph_t *g, const char *path) {
  FILE *f = std::fopen(path, "w");
  if (!f) {
    return -1;
  }
  const bool has_vwgt = !g->vwgt.empty();
  const bool has_ewgt = !g->adjwgt.empty();
  std::fprintf(f, "%u %llu", g->n,
               static_cast<unsigned long long>(g->m / 2));
  if (has_vwgt || has_ewgt) {
    std::fprintf(f, " %d%d", has_vwgt ? 1 : 0, has_ewgt ? 1 : 0);
  }
  std::fputc('\n', f);
  for (u32 u = 0; u < g->n; ++u) {
    bool first = true;
    if (has_vwgt) {
      std::fprintf(f, "%d", g->vwgt[u]);
      first = false;
    }
    for (u32 e = g->xadj[u]; e < g->xadj[u + 1]; ++e) {
      std::fprintf(f, first ? "%u" : " %u", g->adjncy[e] + 1);
      first = false;
      if (has_ewgt) {
        std::fprintf(f, " %d", g->adjwgt[e]);
      }
    }
    std::fputc('\n', f);
  }
  std::fclose(f);
  return 0;
}

// ParHIP binary format (docs/graph_file_format.md "ParHIP Graph File
// Format"; kaminpar-io/parhip_parser.cc:42-136): 24-byte header (version
// bit-field, n, m as u64), then BYTE offsets ((n+1) x EdgeID width,
// relative to the file start), adjacency (m x NodeID width), optional node
// weights, optional edge weights. Version bits (0 = present / 64-bit):
// b0 edge weights absent, b1 node weights absent, b2 edge ids 32-bit,
// b3 node ids 32-bit, b4 node weights 32-bit, b5 edge weights 32-bit.
kmp_graph_t *kmp_read_parhip(const char *path) {
  FILE *f = std::fopen(path, "rb");
  if (!f) {
    std::fprintf(stderr, "kaminpar_amd: cannot open %s\n", path);
    return nullptr;
  }
  u64 header[3];
  if (std::fread(header, 8, 3, f) != 3) {
    std::fclose(f);
    return nullptr;
  }
  const u64 version = header[0];
  const u64 n = header[1];
  const u64 m = header[2];
  const bool has_ewgt = (version & 1) == 0;
  const bool has_vwgt = (version & 2) == 0;
  const int eid_w = (version & 4) == 0 ? 8 : 4;
  const int nid_w = (version & 8) == 0 ? 8 : 4;
  const int vw_w = (version & 16) == 0 ? 8 : 4;
  const int ew_w = (version & 32) == 0 ? 8 : 4;
  if (n > 0xFFFFFFFFull || m > 0xFFFFFFFFull) {
    std::fprintf(stderr, "kaminpar_amd: parhip graph too large for u32 ids\n");
    std::fclose(f);
    return nullptr;
  }

  auto read_ints = [&](int width, u64 count, auto &out) -> bool {
    out.resize(count);
    if (width == 8) {
      std::vector<u64> tmp(count);
      if (std::fread(tmp.data(), 8, count, f) != count) {
        return false;
      }
      for (u64 i = 0; i < count; ++i) {
        out[i] = static_cast<typename std::decay_t<decltype(out)>::value_type>(tmp[i]);
      }
    } else {
      std::vector<u32> tmp(count);
      if (std::fread(tmp.data(), 4, count, f) != count) {
        return false;
      }
      for (u64 i = 0; i < count; ++i) {
        out[i] = static_cast<typename std::decay_t<decltype(out)>::value_type>(tmp[i]);
      }
    }
    return true;
  };

  // offsets are byte addresses; map to edge indices
  const u64 nodes_base = 24 + (n + 1) * static_cast<u64>(eid_w);
  std::vector<u64> off;
  if (!read_ints(eid_w, n + 1, off)) {
    std::fclose(f);
    return nullptr;
  }
  auto *g = new kmp_graph_t();
  g->n = static_cast<u32>(n);
  g->m = m;
  g->xadj.resize(n + 1);
  for (u64 i = 0; i <= n; ++i) {
    g->xadj[i] = static_cast<u32>((off[i] - nodes_base) / nid_w);
  }
  std::vector<u64> adj;
  if (!read_ints(nid_w, m, adj)) {
    delete g;
    std::fclose(f);
    return nullptr;
  }
  g->adjncy.resize(m);
  for (u64 e = 0; e < m; ++e) {
    g->adjncy[e] = static_cast<u32>(adj[e]);
  }
  g->total_node_weight = static_cast<i64>(n);
  if (has_vwgt) {
    std::vector<i64> vw;
    if (!read_ints(vw_w, n, vw)) {
      delete g;
      std::fclose(f);
      return nullptr;
    }
    g->vwgt.resize(n);
    i64 tot = 0;
    for (u64 i = 0; i < n; ++i) {
      g->vwgt[i] = static_cast<i32>(vw[i]);
      tot += vw[i];
    }
    g->total_node_weight = tot;
  }
  if (has_ewgt) {
    std::vector<i64> ew;
    if (!read_ints(ew_w, m, ew)) {
      delete g;
      std::fclose(f);
      return nullptr;
    }
    g->adjwgt.resize(m);
    for (u64 e = 0; e < m; ++e) {
      g->adjwgt[e] = static_cast<i32>(ew[e]);
    }
  }
  std::fclose(f);
  return g;
}

// Writer (32-bit ids and weights), for fixtures and round-trip tests.
int kmp_write_parhip(const kmp_graph_t *g, const char *path) {
  FILE *f = std::fopen(path, "wb");
  if (!f) {
    return -1;
  }
  const bool has_vwgt = !g->vwgt.empty();
  const bool has_ewgt = !g->adjwgt.empty();
  // byte offsets can exceed u32 for large graphs: pick the EdgeID width
  const bool wide_eid = 24 + (static_cast<u64>(g->n) + 1) * 8 + g->m * 4 > 0xFFFFFFFFull;
  // bits set = absent / 32-bit (see reader above)
  const u64 version = (has_ewgt ? 0 : 1) | (has_vwgt ? 0 : 2) |
                      (wide_eid ? 0 : 4) | 8 | 16 | 32;
  u64 header[3] = {version, g->n, g->m};
  std::fwrite(header, 8, 3, f);
  const u64 eid_w = wide_eid ? 8 : 4;
  const u64 nodes_base = 24 + (static_cast<u64>(g->n) + 1) * eid_w;
  if (wide_eid) {
    std::vector<u64> off(g->n + 1);
    for (u32 i = 0; i <= g->n; ++i) {
      off[i] = nodes_base + static_cast<u64>(g->xadj[i]) * 4;
    }
    std::fwrite(off.data(), 8, off.size(), f);
  } else {
    std::vector<u32> off(g->n + 1);
    for (u32 i = 0; i <= g->n; ++i) {
      off[i] = static_cast<u32>(nodes_base + static_cast<u64>(g->xadj[i]) * 4);
    }
    std::fwrite(off.data(), 4, off.size(), f);
  }
  std::fwrite(g->adjncy.data(), 4, g->adjncy.size(), f);
  if (has_vwgt) {
    std::fwrite(g->vwgt.data(), 4, g->vwgt.size(), f);
  }
  if (has_ewgt) {
    std::fwrite(g->adjwgt.data(), 4, g->adjwgt.size(), f);
  }
  std::fclose(f);
  return 0;
}

// Degree-bucket rearrangement: stable counting sort of the vertices by
// exponentially spaced degree buckets, isolated vertices moved to the back
// -- the reference's default NodeOrdering::DEGREE_BUCKETS preprocessing
// (graphutils/permutator.h:30-128 compute_node_permutation_by_degree_buckets
// + build_permuted_graph; bucket(deg) = floor_log2(deg)+1, deg==0 -> last
// bucket, kaminpar-common/degree_buckets.h:20-26). Adjacency-row order is
// preserved; targets are remapped. perm_out[u_old] = u_new (n entries).
// Beyond parity with the reference's preprocessing, this is a locality
// lever for the LP gather path: hub labels become contiguous and stay
// cache-resident.
kmp_graph_t *kmp_rearrange_degree_buckets(const kmp_graph_t *g, u32 *perm_out) {
  const u32 n = g->n;
  constexpr int kBuckets = 33; // 32-bit degrees + deg-0 bucket at the end
  auto bucket_of = [&](u32 u) -> int {
    const u32 deg = g->xadj[u + 1] - g->xadj[u];
    if (deg == 0) {
      return kBuckets - 1;
    }
    return 31 - __builtin_clz(deg) + 1; // floor_log2(deg) + 1
  };

  std::vector<u64> counts(kBuckets + 1, 0);
  for (u32 u = 0; u < n; ++u) {
    ++counts[bucket_of(u) + 1];
  }
  for (int b = 1; b <= kBuckets; ++b) {
    counts[b] += counts[b - 1];
  }
  std::vector<u32> inv(n); // inv[u_new] = u_old
  {
    std::vector<u64> cursor(counts.begin(), counts.end() - 1);
    for (u32 u = 0; u < n; ++u) { // stable within bucket
      const u64 pos = cursor[bucket_of(u)]++;
      perm_out[u] = static_cast<u32>(pos);
      inv[pos] = u;
    }
  }

  auto *out = new kmp_graph_t();
  out->n = n;
  out->m = g->m;
  out->total_node_weight = g->total_node_weight;
  out->xadj.resize(n + 1);
  out->adjncy.resize(g->m);
  if (!g->vwgt.empty()) {
    out->vwgt.resize(n);
  }
  if (!g->adjwgt.empty()) {
    out->adjwgt.resize(g->m);
  }
  out->xadj[0] = 0;
  for (u32 v = 0; v < n; ++v) {
    const u32 u = inv[v];
    out->xadj[v + 1] = out->xadj[v] + (g->xadj[u + 1] - g->xadj[u]);
  }
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 4096)
#endif
  for (long long v = 0; v < static_cast<long long>(n); ++v) {
    const u32 u = inv[v];
    const u64 src = g->xadj[u];
    const u64 dst = out->xadj[v];
    const u32 deg = g->xadj[u + 1] - g->xadj[u];
    for (u32 i = 0; i < deg; ++i) {
      out->adjncy[dst + i] = perm_out[g->adjncy[src + i]];
      if (!g->adjwgt.empty()) {
        out->adjwgt[dst + i] = g->adjwgt[src + i];
      }
    }
    if (!g->vwgt.empty()) {
      out->vwgt[v] = g->vwgt[u];
    }
  }
  return out;
}

} // extern "C"
