// .lux binary IO + synthetic CSC builders + edge-balanced partitioner.
// Format parity: /root/reference/README.md:56-75, tools/converter.cc:108-124.
#include "lux/graph.h"
#include "lux/rmat.h"

#include <cstring>
#include <memory>

namespace lux {

namespace {
bool fread_all(FILE* f, void* p, size_t bytes) {
  return fread(p, 1, bytes, f) == bytes;
}
bool fwrite_all(FILE* f, const void* p, size_t bytes) {
  return fwrite(p, 1, bytes, f) == bytes;
}
struct FileCloser {
  void operator()(FILE* f) const { if (f) fclose(f); }
};
using FilePtr = std::unique_ptr<FILE, FileCloser>;
}  // namespace

bool lux_write(const std::string& path, const HostCSC& g) {
  FilePtr f(fopen(path.c_str(), "wb"));
  if (!f) return false;
  if (!fwrite_all(f.get(), &g.nv, sizeof(V_ID))) return false;
  if (!fwrite_all(f.get(), &g.ne, sizeof(E_ID))) return false;
  if (!fwrite_all(f.get(), g.col_end.data(), sizeof(E_ID) * g.nv)) return false;
  if (!fwrite_all(f.get(), g.src.data(), sizeof(V_ID) * g.ne)) return false;
  if (g.weighted() &&
      !fwrite_all(f.get(), g.weight.data(), sizeof(WeightType) * g.ne))
    return false;
  return true;
}

// Weightedness is not encoded in the header; infer it from the file size the
// way the reference decides by app type (EDGE_WEIGHT compile flag). A file is
// weighted if it is large enough to hold the weight block.
static bool file_is_weighted(FILE* f, V_ID nv, E_ID ne) {
  long cur = ftell(f);
  fseek(f, 0, SEEK_END);
  long sz = ftell(f);
  fseek(f, cur, SEEK_SET);
  uint64_t base = sizeof(V_ID) + sizeof(E_ID) + sizeof(E_ID) * (uint64_t)nv +
                  sizeof(V_ID) * ne;
  return (uint64_t)sz >= base + sizeof(WeightType) * ne;
}

bool lux_read_header(const std::string& path, V_ID* nv, E_ID* ne,
                     bool* weighted) {
  FilePtr f(fopen(path.c_str(), "rb"));
  if (!f) return false;
  if (!fread_all(f.get(), nv, sizeof(V_ID))) return false;
  if (!fread_all(f.get(), ne, sizeof(E_ID))) return false;
  if (weighted) *weighted = file_is_weighted(f.get(), *nv, *ne);
  return true;
}

bool lux_read(const std::string& path, HostCSC* g, bool want_weights) {
  FilePtr f(fopen(path.c_str(), "rb"));
  if (!f) return false;
  if (!fread_all(f.get(), &g->nv, sizeof(V_ID))) return false;
  if (!fread_all(f.get(), &g->ne, sizeof(E_ID))) return false;
  g->col_end.resize(g->nv);
  g->src.resize(g->ne);
  if (!fread_all(f.get(), g->col_end.data(), sizeof(E_ID) * g->nv))
    return false;
  if (!fread_all(f.get(), g->src.data(), sizeof(V_ID) * g->ne)) return false;
  g->weight.clear();
  if (want_weights) {
    if (!file_is_weighted(f.get(), g->nv, g->ne)) return false;
    g->weight.resize(g->ne);
    if (!fread_all(f.get(), g->weight.data(), sizeof(WeightType) * g->ne))
      return false;
  }
  return true;
}

bool lux_read_slice(const std::string& path, V_ID row_left, V_ID row_right,
                    std::vector<E_ID>* col_end, std::vector<V_ID>* src,
                    std::vector<WeightType>* weight) {
  FilePtr f(fopen(path.c_str(), "rb"));
  if (!f) return false;
  V_ID nv;
  E_ID ne;
  if (!fread_all(f.get(), &nv, sizeof(V_ID))) return false;
  if (!fread_all(f.get(), &ne, sizeof(E_ID))) return false;
  if (row_right >= nv || row_left > row_right) return false;
  const uint64_t hdr = sizeof(V_ID) + sizeof(E_ID);
  // Edge range: [col_end[row_left-1], col_end[row_right]).
  E_ID e_lo = 0;
  if (row_left > 0) {
    if (fseeko(f.get(), hdr + sizeof(E_ID) * (uint64_t)(row_left - 1),
               SEEK_SET))
      return false;
    if (!fread_all(f.get(), &e_lo, sizeof(E_ID))) return false;
  } else {
    if (fseeko(f.get(), hdr, SEEK_SET)) return false;
  }
  V_ID nrows = row_right - row_left + 1;
  col_end->resize(nrows);
  if (!fread_all(f.get(), col_end->data(), sizeof(E_ID) * nrows)) return false;
  E_ID e_hi = (*col_end)[nrows - 1];
  src->resize(e_hi - e_lo);
  if (fseeko(f.get(), hdr + sizeof(E_ID) * (uint64_t)nv + sizeof(V_ID) * e_lo,
             SEEK_SET))
    return false;
  if (!fread_all(f.get(), src->data(), sizeof(V_ID) * (e_hi - e_lo)))
    return false;
  if (weight) {
    if (!file_is_weighted(f.get(), nv, ne)) {
      weight->clear();
    } else {
      weight->resize(e_hi - e_lo);
      if (fseeko(f.get(),
                 hdr + sizeof(E_ID) * (uint64_t)nv + sizeof(V_ID) * ne +
                     sizeof(WeightType) * e_lo,
                 SEEK_SET))
        return false;
      if (!fread_all(f.get(), weight->data(),
                     sizeof(WeightType) * (e_hi - e_lo)))
        return false;
    }
  }
  return true;
}

HostCSC edges_to_csc(V_ID nv, const std::vector<V_ID>& src,
                     const std::vector<V_ID>& dst,
                     const std::vector<WeightType>* w) {
  HostCSC g;
  g.nv = nv;
  g.ne = src.size();
  // Counting sort by dst — same algorithm the GPU builder runs (histogram,
  // exclusive scan, scatter), so structures agree up to within-row order.
  std::vector<E_ID> count(nv + 1, 0);
  for (V_ID d : dst) count[d + 1]++;
  for (V_ID v = 0; v < nv; v++) count[v + 1] += count[v];
  g.col_end.resize(nv);
  for (V_ID v = 0; v < nv; v++) g.col_end[v] = count[v + 1];
  g.src.resize(g.ne);
  if (w) g.weight.resize(g.ne);
  std::vector<E_ID> cursor(count.begin(), count.end() - 1);
  for (E_ID e = 0; e < g.ne; e++) {
    E_ID pos = cursor[dst[e]]++;
    g.src[pos] = src[e];
    if (w) g.weight[pos] = (*w)[e];
  }
  return g;
}

HostCSC rmat_csc_cpu(int scale, E_ID ne, uint64_t seed) {
  V_ID nv = (V_ID)1 << scale;
  std::vector<V_ID> s(ne), d(ne);
  for (E_ID e = 0; e < ne; e++) rmat_edge(seed, e, scale, &s[e], &d[e]);
  return edges_to_csc(nv, s, d, nullptr);
}

HostCSC bipartite_csc_cpu(V_ID n_users, V_ID n_items, E_ID ne, uint64_t seed) {
  int item_scale = 0;
  while (((V_ID)1 << item_scale) < n_items) item_scale++;
  std::vector<V_ID> s(ne), d(ne);
  std::vector<WeightType> w(ne);
  for (E_ID e = 0; e < ne; e++) {
    bipartite_edge(seed, e, n_users, n_items, item_scale, &s[e], &d[e]);
    w[e] = rmat_weight(seed, e >> 1);  // both directions share the rating
  }
  return edges_to_csc(n_users + n_items, s, d, &w);
}

Partition partition_edge_balanced(V_ID nv, E_ID ne, const E_ID* col_end,
                                  int nparts) {
  // Greedy contiguous cut at the edge midpoints: part p covers vertices
  // until its cumulative edges reach (p+1) * ceil(ne/nparts) (the
  // reference's edge_cap rule, core/pull_model.inl:108-131).
  Partition part;
  part.row_left.assign(nparts, 0);
  part.row_right.assign(nparts, 0);
  part.col_left.assign(nparts, 0);
  part.col_right.assign(nparts, 0);
  E_ID edge_cap = (ne + nparts - 1) / nparts;
  V_ID v = 0;
  for (int p = 0; p < nparts; p++) {
    part.row_left[p] = v;
    part.col_left[p] = v == 0 ? 0 : col_end[v - 1];
    E_ID target = edge_cap * (E_ID)(p + 1);
    if (target > ne) target = ne;
    while (v < nv && (p == nparts - 1 || col_end[v] <= target ||
                      v == part.row_left[p]))
      v++;
    part.row_right[p] = v == part.row_left[p] ? v : v - 1;  // empty guard
    if (v == part.row_left[p]) {
      // Empty partition (nv exhausted): mark with left > right.
      part.row_left[p] = 1;
      part.row_right[p] = 0;
      part.col_left[p] = part.col_right[p] = ne;
    } else {
      part.col_right[p] = col_end[part.row_right[p]];
    }
  }
  return part;
}

}  // namespace lux
