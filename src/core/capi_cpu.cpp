// C ABI for the CPU core, consumed by lux_amd/_native.py via ctypes.
// All pointers are caller-allocated (numpy / torch CPU tensors).
#include <cstdio>
#include <cstring>
#include <vector>

#include "lux/graph.h"
#include "lux/rmat.h"

namespace lux {
// engines.cpp
void pagerank_init(V_ID nv, const V_ID* out_deg, float* pr);
void pagerank_iter_part(V_ID nv, V_ID row_left, V_ID row_right, E_ID col_left,
                        const E_ID* col_end, const V_ID* src,
                        const V_ID* out_deg, const float* old_pr,
                        float* new_pr_part);
void out_degrees(V_ID nv, E_ID ne, const V_ID* src, V_ID* deg);
void pagerank_cpu(const HostCSC& g, int iters, float* pr_out);
V_ID sssp_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                    const E_ID* col_end, const V_ID* src,
                    const V_ID* old_label, V_ID* new_label_part);
V_ID cc_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                  const E_ID* col_end, const V_ID* src, const V_ID* old_label,
                  V_ID* new_label_part);
int sssp_cpu(const HostCSC& g, V_ID source, V_ID* label_out);
int cc_cpu(const HostCSC& g, V_ID* label_out);
E_ID sssp_check(const HostCSC& g, const V_ID* label);
E_ID cc_check(const HostCSC& g, const V_ID* label);
void cf_init(V_ID nv, int K, float* vec);
void cf_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                  const E_ID* col_end, const V_ID* src, const WeightType* w,
                  int K, const float* old_vec, float* new_vec_part);
void cf_cpu(const HostCSC& g, int K, int iters, float* vec_out);
double cf_loss(const HostCSC& g, int K, const float* vec);
}  // namespace lux

using namespace lux;

namespace {
HostCSC view_csc(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                 const uint32_t* src, const int32_t* weight) {
  HostCSC g;
  g.nv = nv;
  g.ne = ne;
  g.col_end.assign(col_end, col_end + nv);
  g.src.assign(src, src + ne);
  if (weight) g.weight.assign(weight, weight + ne);
  return g;
}
}  // namespace

extern "C" {

// ---- generation ----
void lux_rmat_edges(uint64_t seed, int scale, uint64_t ne, uint32_t* src,
                    uint32_t* dst) {
  for (uint64_t e = 0; e < ne; e++) rmat_edge(seed, e, scale, &src[e], &dst[e]);
}
void lux_rmat_edges_folded(uint64_t seed, int scale, uint32_t nv,
                           uint64_t ne, uint32_t* src, uint32_t* dst) {
  for (uint64_t e = 0; e < ne; e++)
    rmat_edge_folded(seed, e, scale, nv, &src[e], &dst[e]);
}
void lux_bipartite_edges(uint64_t seed, uint32_t n_users, uint32_t n_items,
                         uint64_t ne, uint32_t* src, uint32_t* dst,
                         int32_t* w) {
  int item_scale = 0;
  while (((uint32_t)1 << item_scale) < n_items) item_scale++;
  for (uint64_t e = 0; e < ne; e++) {
    bipartite_edge(seed, e, n_users, n_items, item_scale, &src[e], &dst[e]);
    w[e] = rmat_weight(seed, e >> 1);  // both directions share the rating
  }
}
void lux_edges_to_csc(uint32_t nv, uint64_t ne, const uint32_t* src,
                      const uint32_t* dst, const int32_t* w,
                      uint64_t* col_end, uint32_t* out_src, int32_t* out_w) {
  std::vector<V_ID> sv(src, src + ne), dv(dst, dst + ne);
  std::vector<WeightType> wv;
  if (w) wv.assign(w, w + ne);
  HostCSC g = edges_to_csc(nv, sv, dv, w ? &wv : nullptr);
  std::memcpy(col_end, g.col_end.data(), sizeof(uint64_t) * nv);
  std::memcpy(out_src, g.src.data(), sizeof(uint32_t) * ne);
  if (w && out_w) std::memcpy(out_w, g.weight.data(), sizeof(int32_t) * ne);
}

// ---- .lux IO ----
int lux_io_write(const char* path, uint32_t nv, uint64_t ne,
                 const uint64_t* col_end, const uint32_t* src,
                 const int32_t* weight) {
  HostCSC g = view_csc(nv, ne, col_end, src, weight);
  return lux_write(path, g) ? 0 : -1;
}
int lux_io_read_header(const char* path, uint32_t* nv, uint64_t* ne,
                       int* weighted) {
  bool w = false;
  if (!lux_read_header(path, nv, ne, &w)) return -1;
  *weighted = w ? 1 : 0;
  return 0;
}
int lux_io_read(const char* path, uint64_t* col_end, uint32_t* src,
                int32_t* weight) {
  HostCSC g;
  if (!lux_read(path, &g, weight != nullptr)) return -1;
  std::memcpy(col_end, g.col_end.data(), sizeof(uint64_t) * g.nv);
  std::memcpy(src, g.src.data(), sizeof(uint32_t) * g.ne);
  if (weight)
    std::memcpy(weight, g.weight.data(), sizeof(int32_t) * g.ne);
  return 0;
}

// Per-partition slice read (reference parity: the per-node fseeko load,
// core/push_model.inl:100-119): reads ONLY [row_left, row_right]'s col_end
// slice + edge slice. Caller sizes src/weight from a prior col_end read
// (lux_io_read_col_end); returns the slice edge count or -1.
int lux_io_read_col_end(const char* path, uint64_t* col_end) {
  uint32_t nv;
  uint64_t ne;
  bool w;
  if (!lux_read_header(path, &nv, &ne, &w)) return -1;
  FILE* f = fopen(path, "rb");
  if (!f) return -1;
  if (fseeko(f, sizeof(uint32_t) + sizeof(uint64_t), SEEK_SET)) {
    fclose(f);
    return -1;
  }
  size_t got = fread(col_end, sizeof(uint64_t), nv, f);
  fclose(f);
  return got == nv ? 0 : -1;
}

int64_t lux_io_read_slice(const char* path, uint32_t row_left,
                          uint32_t row_right, uint64_t* col_end_slice,
                          uint32_t* src, int32_t* weight) {
  std::vector<E_ID> ce;
  std::vector<V_ID> s;
  std::vector<WeightType> w;
  if (!lux_read_slice(path, row_left, row_right, &ce, &s,
                      weight ? &w : nullptr))
    return -1;
  if (weight && w.empty()) return -1;  // asked for weights, file has none
  std::memcpy(col_end_slice, ce.data(), sizeof(uint64_t) * ce.size());
  std::memcpy(src, s.data(), sizeof(uint32_t) * s.size());
  if (weight) std::memcpy(weight, w.data(), sizeof(int32_t) * w.size());
  return (int64_t)s.size();
}

// ---- partitioner ----
void lux_partition(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                   int nparts, uint32_t* row_left, uint32_t* row_right,
                   uint64_t* col_left, uint64_t* col_right) {
  Partition p = partition_edge_balanced(nv, ne, col_end, nparts);
  for (int i = 0; i < nparts; i++) {
    row_left[i] = p.row_left[i];
    row_right[i] = p.row_right[i];
    col_left[i] = p.col_left[i];
    col_right[i] = p.col_right[i];
  }
}

// ---- engines ----
void lux_out_degrees(uint32_t nv, uint64_t ne, const uint32_t* src,
                     uint32_t* deg) {
  out_degrees(nv, ne, src, deg);
}
void lux_pagerank_init(uint32_t nv, const uint32_t* deg, float* pr) {
  pagerank_init(nv, deg, pr);
}
void lux_pagerank_iter_part(uint32_t nv, uint32_t row_left, uint32_t row_right,
                            uint64_t col_left, const uint64_t* col_end,
                            const uint32_t* src, const uint32_t* deg,
                            const float* old_pr, float* new_pr_part) {
  pagerank_iter_part(nv, row_left, row_right, col_left, col_end, src, deg,
                     old_pr, new_pr_part);
}
void lux_pagerank_cpu(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                      const uint32_t* src, int iters, float* pr_out) {
  HostCSC g = view_csc(nv, ne, col_end, src, nullptr);
  pagerank_cpu(g, iters, pr_out);
}
uint32_t lux_sssp_iter_part(uint32_t row_left, uint32_t row_right,
                            uint64_t col_left, const uint64_t* col_end,
                            const uint32_t* src, const uint32_t* old_label,
                            uint32_t* new_label_part) {
  return sssp_iter_part(row_left, row_right, col_left, col_end, src,
                        old_label, new_label_part);
}
uint32_t lux_cc_iter_part(uint32_t row_left, uint32_t row_right,
                          uint64_t col_left, const uint64_t* col_end,
                          const uint32_t* src, const uint32_t* old_label,
                          uint32_t* new_label_part) {
  return cc_iter_part(row_left, row_right, col_left, col_end, src, old_label,
                      new_label_part);
}
int lux_sssp_cpu(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                 const uint32_t* src, uint32_t source, uint32_t* label_out) {
  HostCSC g = view_csc(nv, ne, col_end, src, nullptr);
  return sssp_cpu(g, source, label_out);
}
int lux_cc_cpu(uint32_t nv, uint64_t ne, const uint64_t* col_end,
               const uint32_t* src, uint32_t* label_out) {
  HostCSC g = view_csc(nv, ne, col_end, src, nullptr);
  return cc_cpu(g, label_out);
}
uint64_t lux_sssp_check(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                        const uint32_t* src, const uint32_t* label) {
  HostCSC g = view_csc(nv, ne, col_end, src, nullptr);
  return sssp_check(g, label);
}
uint64_t lux_cc_check(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                      const uint32_t* src, const uint32_t* label) {
  HostCSC g = view_csc(nv, ne, col_end, src, nullptr);
  return cc_check(g, label);
}
void lux_cf_init(uint32_t nv, int K, float* vec) { cf_init(nv, K, vec); }
void lux_cf_iter_part(uint32_t row_left, uint32_t row_right,
                      uint64_t col_left, const uint64_t* col_end,
                      const uint32_t* src, const int32_t* w, int K,
                      const float* old_vec, float* new_vec_part) {
  cf_iter_part(row_left, row_right, col_left, col_end, src, w, K, old_vec,
               new_vec_part);
}
void lux_cf_cpu(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                const uint32_t* src, const int32_t* w, int K, int iters,
                float* vec_out) {
  HostCSC g = view_csc(nv, ne, col_end, src, w);
  cf_cpu(g, K, iters, vec_out);
}
double lux_cf_loss(uint32_t nv, uint64_t ne, const uint64_t* col_end,
                   const uint32_t* src, const int32_t* w, int K,
                   const float* vec) {
  HostCSC g = view_csc(nv, ne, col_end, src, w);
  return cf_loss(g, K, vec);
}

}  // extern "C"
