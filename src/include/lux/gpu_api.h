// C ABI of liblux_gpu.so (gfx950 kernels). Consumed by the Python ctypes
// bindings (lux_amd/_native_gpu.py) and the native C++ runtime
// (src/runtime/). Pointers are device pointers; `stream` is a hipStream_t.
#pragma once
#include <cstdint>

#include "types.h"

// HIP vector type fwd-compat for non-HIP translation units
#ifndef __HIP_PLATFORM_AMD__
struct lux_uint2 { uint32_t x, y; };
#else
using lux_uint2 = uint2;
#endif

extern "C" {

// builder.hip
void lux_gpu_rmat_edges(uint64_t stream, uint64_t seed, int scale,
                        uint64_t ne, lux::V_ID* src, lux::V_ID* dst);
void lux_gpu_rmat_edges_folded(uint64_t stream, uint64_t seed, int scale,
                               lux::V_ID nv, uint64_t ne, lux::V_ID* src,
                               lux::V_ID* dst);
void lux_gpu_bipartite_edges(uint64_t stream, uint64_t seed,
                             lux::V_ID n_users, lux::V_ID n_items,
                             uint64_t ne, lux::V_ID* src, lux::V_ID* dst,
                             lux::WeightType* w);
void lux_gpu_hist_u32(uint64_t stream, uint64_t n, const lux::V_ID* ids,
                      uint32_t* hist);
void lux_gpu_rmat_edges_chunk(uint64_t stream, uint64_t seed, int scale,
                              uint64_t e0, uint64_t ne, lux::V_ID* src,
                              lux::V_ID* dst);
void lux_gpu_rmat_edges_folded_chunk(uint64_t stream, uint64_t seed,
                                     int scale, lux::V_ID nv, uint64_t e0,
                                     uint64_t ne, lux::V_ID* src,
                                     lux::V_ID* dst);
void lux_gpu_bipartite_edges_chunk(uint64_t stream, uint64_t seed,
                                   lux::V_ID n_users, lux::V_ID n_items,
                                   uint64_t e0, uint64_t ne, lux::V_ID* src,
                                   lux::V_ID* dst, lux::WeightType* w);
void lux_gpu_slice_scatter(uint64_t stream, uint64_t n, const lux::V_ID* src,
                           const lux::V_ID* dst, const lux::WeightType* w,
                           lux::V_ID rl, lux::V_ID rr,
                           unsigned long long* cursor, lux::V_ID* out_col,
                           lux::WeightType* out_w);
uint32_t lux_gpu_scan_partials_size(uint64_t n);
void lux_gpu_scan_end_offsets(uint64_t stream, uint64_t n,
                              const uint32_t* in, lux::E_ID* out_end,
                              unsigned long long* partials);
void lux_gpu_u64_to_u32(uint64_t stream, uint64_t n,
                        const unsigned long long* in, uint32_t* out);
void lux_gpu_edges_to_csc(uint64_t stream, uint32_t nv, uint64_t ne,
                          const lux::V_ID* src, const lux::V_ID* dst,
                          const lux::WeightType* w, lux::E_ID* col_end,
                          lux::V_ID* out_src, lux::WeightType* out_w,
                          uint32_t* hist, unsigned long long* cursor,
                          unsigned long long* partials);
void lux_gpu_local_row_ptr(uint64_t stream, uint32_t vp, lux::E_ID col_left,
                           const lux::E_ID* col_end_slice,
                           lux::E_ID* row_ptr_loc);
void lux_gpu_blocked_count(uint64_t stream, uint64_t ep, const lux::V_ID* col,
                           const lux::E_ID* row_ptr_loc, lux::V_ID vp,
                           const lux::V_ID* bounds, int nb, lux::V_ID lo,
                           lux::V_ID hi, uint32_t* counts);
void lux_gpu_blocked_scatter(uint64_t stream, uint64_t ep,
                             const lux::V_ID* col,
                             const lux::E_ID* row_ptr_loc, lux::V_ID vp,
                             const lux::V_ID* bounds, int nb, lux::V_ID lo,
                             lux::V_ID hi, unsigned long long* cursor,
                             lux::V_ID* out_col);

// pull.hip
void lux_gpu_build_bins(uint64_t stream, uint32_t vp,
                        const lux::E_ID* row_ptr, lux::V_ID* bin0,
                        lux::V_ID* bin1, lux_uint2* bin2, lux::V_ID* bin2v,
                        uint32_t* counters);
void lux_gpu_pull_iter(uint64_t stream, int mode, uint32_t n0,
                       const lux::V_ID* bin0, uint32_t n1,
                       const lux::V_ID* bin1, uint32_t n2,
                       const lux_uint2* bin2, uint32_t nbig,
                       const lux::V_ID* bin2v, const void* row_ptr,
                       int row_u32, const lux::V_ID* col, const void* oldv,
                       void* newv, const lux::V_ID* deg, lux::V_ID row_left,
                       float init_rank);
void lux_gpu_pull_finish_pr(uint64_t stream, lux::V_ID vp, float* newv,
                            const lux::V_ID* deg, lux::V_ID row_left,
                            float init_rank);

// push.hip
void lux_gpu_csr_scatter(uint64_t stream, uint64_t ep, const lux::V_ID* col,
                         const lux::E_ID* row_ptr_loc, lux::V_ID vp,
                         lux::V_ID row_left, unsigned long long* cursor,
                         lux::V_ID* push_col);
void lux_gpu_frontier_expand(uint64_t stream, int old_dense,
                             lux::V_ID in_row_left, lux::V_ID in_count,
                             const uint8_t* old_seg,
                             const uint32_t* qlabels /*nullable*/,
                             uint32_t* labels_repair /*nullable*/,
                             const lux::E_ID* push_row_ptr,
                             lux_uint2* items, uint32_t* counter,
                             uint32_t max_items);
void lux_gpu_frontier_expand_auto(uint64_t stream, lux::V_ID verts,
                                  lux::V_ID in_row_left, const uint8_t* seg,
                                  const uint32_t* qlabels,
                                  uint32_t* labels_repair,
                                  const lux::E_ID* push_row_ptr,
                                  lux_uint2* items, uint32_t* counter,
                                  uint32_t max_items);
void lux_gpu_frontier_fixup(uint64_t stream, lux::V_ID vp,
                            lux::V_ID row_left, lux::V_ID capacity,
                            int built_dense, const uint32_t* snapshot,
                            const uint32_t* labels_part,
                            const uint32_t* deg_part, uint8_t* new_seg,
                            uint32_t* annex, uint8_t* tmp_seg,
                            uint32_t* meta,
                            const uint32_t* item_counter /*nullable*/,
                            uint32_t max_items);
void lux_gpu_push_chunk_scatter(uint64_t stream, int is_min, int new_dense,
                                const lux_uint2* items,
                                const uint32_t* counter, uint32_t max_items,
                                const lux::E_ID* push_row_ptr,
                                const lux::V_ID* push_col,
                                const uint32_t* old_labels,
                                const uint32_t* snapshot,
                                uint32_t* new_labels, lux::V_ID my_row_left,
                                uint8_t* new_seg, lux::V_ID capacity,
                                uint32_t* visited_bits /*nullable*/);
void lux_gpu_bits_from_labels(uint64_t stream, lux::V_ID vp,
                              const uint32_t* labels, uint32_t* bits);

void lux_gpu_build_bitmap(uint64_t stream, lux::V_ID vp,
                          const uint32_t* snapshot,
                          const uint32_t* new_labels, uint8_t* seg);
void lux_gpu_d2s(uint64_t stream, lux::V_ID vp, lux::V_ID row_left,
                 const uint8_t* dense_seg, uint8_t* sparse_seg);
void lux_gpu_publish_labels_guarded(uint64_t stream, lux::V_ID vp,
                                    const uint32_t* meta,
                                    const uint32_t* labels_part,
                                    uint32_t* labels_slice);
void lux_gpu_check(uint64_t stream, int is_min, lux::V_ID vp,
                   lux::V_ID row_left, const lux::E_ID* row_ptr_loc,
                   const lux::V_ID* col, const uint32_t* labels,
                   unsigned long long* mistakes);

// cf.hip
void lux_gpu_cf_seed(uint64_t stream, uint64_t n, const float* oldv,
                     float* newv);
void lux_gpu_cf_iter(uint64_t stream, uint32_t n0, const lux::V_ID* bin0,
                     uint32_t n1, const lux::V_ID* bin1, uint32_t n2,
                     const lux_uint2* bin2, uint32_t nbig,
                     const lux::V_ID* bin2v, const lux::E_ID* row_ptr,
                     const lux::V_ID* col, const lux::WeightType* w,
                     const float* oldv, float* newv, lux::V_ID row_left,
                     int K);

// cc_uf.hip
void lux_gpu_count_diff(uint64_t stream, uint64_t n, const uint32_t* a,
                        const uint32_t* b, unsigned long long* out);
void lux_gpu_uf_union_binned(uint64_t stream, uint32_t n0,
                             const lux::V_ID* bin0, uint32_t n1,
                             const lux::V_ID* bin1, uint32_t n2,
                             const lux_uint2* bin2, const lux::E_ID* row_ptr,
                             const lux::V_ID* col, lux::V_ID row_left,
                             lux::V_ID* parent, const uint32_t* gbits);
void lux_gpu_uf_union_kth(uint64_t stream, lux::V_ID vp,
                          const lux::E_ID* row_ptr, const lux::V_ID* col,
                          lux::V_ID row_left, lux::V_ID* parent, uint32_t k);
void lux_gpu_cc_giant_bits(uint64_t stream, lux::V_ID nv,
                           const lux::V_ID* labels, lux::V_ID giant,
                           uint32_t* bits);
void lux_gpu_uf_union_star(uint64_t stream, lux::V_ID nv,
                           const lux::V_ID* star, lux::V_ID* parent);
void lux_gpu_uf_flatten(uint64_t stream, lux::V_ID nv, lux::V_ID* parent,
                        lux::V_ID* labels);

// cf_als.hip
void lux_gpu_cf_als_iter(uint64_t stream, uint32_t n0, const lux::V_ID* bin0,
                         uint32_t n1, const lux::V_ID* bin1, uint32_t n2,
                         const lux_uint2* bin2, uint32_t nbig,
                         const lux::V_ID* bin2v, const int* hubidx,
                         float* gram_scratch, float* rhs_scratch,
                         const lux::E_ID* row_ptr, const lux::V_ID* col,
                         const lux::WeightType* w, const float* oldv,
                         const uint16_t* oldv_bf /*nullable*/, float* newv,
                         lux::V_ID row_left, int K);

}  // extern "C"
