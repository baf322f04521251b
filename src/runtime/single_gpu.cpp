#include "single_gpu.h"

#include <algorithm>
#include <cassert>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace lux {

DeviceArena::DeviceArena(size_t bytes) : cap_(bytes) {
  LUX_OK(hipMalloc(&base_, bytes));
}
DeviceArena::~DeviceArena() { hipFree(base_); }
void* DeviceArena::alloc(size_t bytes) {
  size_t off = (used_ + 255) & ~size_t(255);
  if (off + bytes > cap_) {
    fprintf(stderr, "lux: FB arena exhausted (%zu + %zu > %zu)\n", off,
            bytes, cap_);
    abort();
  }
  used_ = off + bytes;
  return base_ + off;
}

DeviceGraph DeviceGraph::upload(const HostCSC& g, DeviceArena& arena,
                                hipStream_t s) {
  DeviceGraph d;
  d.nv = g.nv;
  d.ne = g.ne;
  d.col_end = arena.alloc_n<E_ID>(g.nv);
  d.src = arena.alloc_n<V_ID>(g.ne);
  LUX_OK(hipMemcpyAsync(d.col_end, g.col_end.data(),
                           sizeof(E_ID) * g.nv, hipMemcpyHostToDevice, s));
  LUX_OK(hipMemcpyAsync(d.src, g.src.data(), sizeof(V_ID) * g.ne,
                           hipMemcpyHostToDevice, s));
  if (g.weighted()) {
    d.weight = arena.alloc_n<WeightType>(g.ne);
    LUX_OK(hipMemcpyAsync(d.weight, g.weight.data(),
                             sizeof(WeightType) * g.ne,
                             hipMemcpyHostToDevice, s));
  }
  LUX_OK(hipStreamSynchronize(s));
  return d;
}

DeviceGraph DeviceGraph::rmat(int scale, E_ID ne, uint64_t seed,
                              DeviceArena& arena, hipStream_t s) {
  DeviceGraph d;
  d.nv = (V_ID)1 << scale;
  d.ne = ne;
  d.col_end = arena.alloc_n<E_ID>(d.nv);
  d.src = arena.alloc_n<V_ID>(ne);
  // temporaries via hipMalloc (freed after build, not arena-held)
  V_ID *esrc, *edst;
  uint32_t* hist;
  unsigned long long *cursor, *partials;
  LUX_OK(hipMalloc(&esrc, sizeof(V_ID) * ne));
  LUX_OK(hipMalloc(&edst, sizeof(V_ID) * ne));
  LUX_OK(hipMalloc(&hist, sizeof(uint32_t) * d.nv));
  LUX_OK(hipMalloc(&cursor, sizeof(uint64_t) * d.nv));
  LUX_OK(
      hipMalloc(&partials, sizeof(uint64_t) * lux_gpu_scan_partials_size(d.nv)));
  LUX_OK(hipMemsetAsync(hist, 0, sizeof(uint32_t) * d.nv, s));
  lux_gpu_rmat_edges((uint64_t)s, seed, scale, ne, esrc, edst);
  lux_gpu_edges_to_csc((uint64_t)s, d.nv, ne, esrc, edst, nullptr, d.col_end,
                       d.src, nullptr, hist, cursor, partials);
  LUX_OK(hipStreamSynchronize(s));
  hipFree(esrc);
  hipFree(edst);
  hipFree(hist);
  hipFree(cursor);
  hipFree(partials);
  return d;
}

DeviceGraph DeviceGraph::bipartite(V_ID n_users, V_ID n_items, E_ID ne,
                                   uint64_t seed, DeviceArena& arena,
                                   hipStream_t s) {
  DeviceGraph d;
  d.nv = n_users + n_items;
  d.ne = ne;
  d.col_end = arena.alloc_n<E_ID>(d.nv);
  d.src = arena.alloc_n<V_ID>(ne);
  d.weight = arena.alloc_n<WeightType>(ne);
  V_ID *esrc, *edst;
  WeightType* ew;
  uint32_t* hist;
  unsigned long long *cursor, *partials;
  LUX_OK(hipMalloc(&esrc, sizeof(V_ID) * ne));
  LUX_OK(hipMalloc(&edst, sizeof(V_ID) * ne));
  LUX_OK(hipMalloc(&ew, sizeof(WeightType) * ne));
  LUX_OK(hipMalloc(&hist, sizeof(uint32_t) * d.nv));
  LUX_OK(hipMalloc(&cursor, sizeof(uint64_t) * d.nv));
  LUX_OK(hipMalloc(&partials,
                   sizeof(uint64_t) * lux_gpu_scan_partials_size(d.nv)));
  LUX_OK(hipMemsetAsync(hist, 0, sizeof(uint32_t) * d.nv, s));
  lux_gpu_bipartite_edges((uint64_t)s, seed, n_users, n_items, ne, esrc,
                          edst, ew);
  lux_gpu_edges_to_csc((uint64_t)s, d.nv, ne, esrc, edst, ew, d.col_end,
                       d.src, d.weight, hist, cursor, partials);
  LUX_OK(hipStreamSynchronize(s));
  hipFree(esrc);
  hipFree(edst);
  hipFree(ew);
  hipFree(hist);
  hipFree(cursor);
  hipFree(partials);
  return d;
}

void Bins::build(const E_ID* row_ptr_loc, V_ID vp, E_ID ep,
                 DeviceArena& arena, hipStream_t s) {
  bin0 = arena.alloc_n<V_ID>(vp);
  bin1 = arena.alloc_n<V_ID>(vp);
  uint32_t nbig_max = (uint32_t)std::min<uint64_t>(vp, ep / 2048 + 1);
  uint32_t n2_max = (uint32_t)(ep / 8192 + nbig_max + 1);
  bin2 = arena.alloc_n<lux_uint2>(n2_max);
  bin2v = arena.alloc_n<V_ID>(nbig_max);
  uint32_t* counters;
  LUX_OK(hipMalloc(&counters, 4 * sizeof(uint32_t)));
  LUX_OK(hipMemsetAsync(counters, 0, 16, s));
  lux_gpu_build_bins((uint64_t)s, vp, row_ptr_loc, bin0, bin1, bin2, bin2v,
                     counters);
  uint32_t c[4];
  LUX_OK(hipMemcpyAsync(c, counters, 16, hipMemcpyDeviceToHost, s));
  LUX_OK(hipStreamSynchronize(s));
  hipFree(counters);
  n0 = c[0];
  n1 = c[1];
  n2 = c[2];
  nbig = c[3];
}

// ---------------- src-blocked pull (BlockedPull) ----------------

static int blocked_shift(V_ID nv) {
  // LUX_NATIVE_BLOCK_SHIFT forces blocking at any nv (tests); default:
  // 32 MB windows, only when the gather window exceeds the 256 MiB LLC
  if (const char* e = getenv("LUX_NATIVE_BLOCK_SHIFT")) return atoi(e);
  return (uint64_t)nv * 4 > (256ull << 20) ? 23 : 0;
}

size_t BlockedPull::arena_bytes(V_ID nv, V_ID vp, E_ID ep) {
  int shift = blocked_shift(nv);
  if (!shift || ep == 0) return 0;
  uint64_t sb = ((uint64_t)nv + (1u << shift) - 1) >> shift;
  // blk_col + per-block u32 row tables + compacted bins (each (block,row)
  // pair with edges lands in exactly one bin list => <= ep entries) +
  // bin2 chunk lists + slack
  return 4ull * ep + sb * 4ull * (vp + 1) + 5ull * ep + (64ull << 20);
}

void BlockedPull::build(const E_ID* row_ptr_loc, const V_ID* col, V_ID vp_,
                        E_ID ep, V_ID nv, DeviceArena& arena,
                        hipStream_t s) {
  vp = vp_;
  int shift = blocked_shift(nv);
  if (!shift || ep == 0 || vp == 0) return;
  std::vector<V_ID> hb;
  for (uint64_t b = 0; b < nv; b += (1u << shift)) hb.push_back((V_ID)b);
  hb.push_back(nv);
  int sb = (int)hb.size() - 1;
  if (sb <= 1) return;
  V_ID* bounds;
  LUX_OK(hipMalloc(&bounds, sizeof(V_ID) * hb.size()));
  LUX_OK(hipMemcpyAsync(bounds, hb.data(), sizeof(V_ID) * hb.size(),
                        hipMemcpyHostToDevice, s));
  uint64_t n = (uint64_t)sb * vp;
  uint32_t* counts;
  unsigned long long *cursor, *partials;
  LUX_OK(hipMalloc(&counts, 4ull * n));
  LUX_OK(hipMalloc(&cursor, 8ull * (n + 1)));
  LUX_OK(hipMalloc(&partials,
                   8ull * lux_gpu_scan_partials_size(n)));
  LUX_OK(hipMemsetAsync(counts, 0, 4ull * n, s));
  LUX_OK(hipMemsetAsync(cursor, 0, 8, s));
  lux_gpu_blocked_count((uint64_t)s, ep, col, row_ptr_loc, vp, bounds,
                        sb + 1, 0, nv, counts);
  lux_gpu_scan_end_offsets((uint64_t)s, n, counts, (E_ID*)cursor + 1,
                           partials);
  V_ID* blk_col = arena.alloc_n<V_ID>(ep);
  lux_gpu_blocked_scatter((uint64_t)s, ep, col, row_ptr_loc, vp, bounds,
                          sb + 1, 0, nv, cursor, blk_col);
  // post-scatter, cursor[i] == end offset of slot i
  E_ID* row64;
  LUX_OK(hipMalloc(&row64, 8ull * (vp + 1)));
  // temp (uncompacted) bin buffers reused across blocks
  V_ID *t0, *t1, *t2v;
  lux_uint2* t2;
  uint32_t* counters;
  uint32_t nbig_max = (uint32_t)std::min<uint64_t>(vp, ep / 2048 + 1);
  uint32_t n2_max = (uint32_t)(ep / 8192 + nbig_max + 1);
  LUX_OK(hipMalloc(&t0, 4ull * vp));
  LUX_OK(hipMalloc(&t1, 4ull * vp));
  LUX_OK(hipMalloc(&t2, 8ull * n2_max));
  LUX_OK(hipMalloc(&t2v, 4ull * nbig_max));
  LUX_OK(hipMalloc(&counters, 16));
  uint64_t begin = 0;
  for (int b = 0; b < sb; b++) {
    // post-scatter, cursor[i] (i < n) holds slot i's END offset
    unsigned long long end;
    LUX_OK(hipMemcpyAsync(&end, cursor + ((uint64_t)(b + 1) * vp - 1), 8,
                          hipMemcpyDeviceToHost, s));
    LUX_OK(hipStreamSynchronize(s));
    if (end == begin) continue;
    lux_gpu_local_row_ptr((uint64_t)s, vp, begin,
                          (E_ID*)(cursor + (uint64_t)b * vp), row64);
    LUX_OK(hipMemsetAsync(counters, 0, 16, s));
    lux_gpu_build_bins((uint64_t)s, vp, row64, t0, t1, t2, t2v, counters);
    uint32_t hc[4];
    LUX_OK(hipMemcpyAsync(hc, counters, 16, hipMemcpyDeviceToHost, s));
    LUX_OK(hipStreamSynchronize(s));
    Blk blk;
    blk.n0 = hc[0];
    blk.n1 = hc[1];
    blk.n2 = hc[2];
    blk.nbig = hc[3];
    blk.bin0 = arena.alloc_n<V_ID>(hc[0] ? hc[0] : 1);
    blk.bin1 = arena.alloc_n<V_ID>(hc[1] ? hc[1] : 1);
    blk.bin2 = arena.alloc_n<lux_uint2>(hc[2] ? hc[2] : 1);
    blk.bin2v = arena.alloc_n<V_ID>(hc[3] ? hc[3] : 1);
    if (hc[0])
      LUX_OK(hipMemcpyAsync(blk.bin0, t0, 4ull * hc[0],
                            hipMemcpyDeviceToDevice, s));
    if (hc[1])
      LUX_OK(hipMemcpyAsync(blk.bin1, t1, 4ull * hc[1],
                            hipMemcpyDeviceToDevice, s));
    if (hc[2])
      LUX_OK(hipMemcpyAsync(blk.bin2, t2, 8ull * hc[2],
                            hipMemcpyDeviceToDevice, s));
    if (hc[3])
      LUX_OK(hipMemcpyAsync(blk.bin2v, t2v, 4ull * hc[3],
                            hipMemcpyDeviceToDevice, s));
    blk.row32 = arena.alloc_n<uint32_t>(vp + 1);
    lux_gpu_u64_to_u32((uint64_t)s, (uint64_t)vp + 1,
                       (const unsigned long long*)row64, blk.row32);
    blk.col = blk_col + begin;
    blocks.push_back(blk);
    begin = end;
  }
  LUX_OK(hipStreamSynchronize(s));
  hipFree(bounds);
  hipFree(counts);
  hipFree(cursor);
  hipFree(partials);
  hipFree(row64);
  hipFree(t0);
  hipFree(t1);
  hipFree(t2);
  hipFree(t2v);
  hipFree(counters);
}

void BlockedPull::sweep(int mode, const void* oldv, void* newv,
                        const V_ID* deg, V_ID row_left, float init_rank,
                        hipStream_t s) const {
  for (const Blk& b : blocks)
    lux_gpu_pull_iter((uint64_t)s, mode, b.n0, b.bin0, b.n1, b.bin1, b.n2,
                      b.bin2, b.nbig, b.bin2v, b.row32, 1, b.col, oldv,
                      newv, deg, row_left, init_rank);
}

// ---------------- PageRank ----------------

SingleGpuPagerank::SingleGpuPagerank(const DeviceGraph& g, DeviceArena& arena,
                                     hipStream_t s)
    : g_(g), s_(s) {
  row_ptr_ = arena.alloc_n<E_ID>(g.nv + 1);
  lux_gpu_local_row_ptr((uint64_t)s, g.nv, 0, g.col_end, row_ptr_);
  bins_.build(row_ptr_, g.nv, g.ne, arena, s);
  blocked_.build(row_ptr_, g.src, g.nv, g.ne, g.nv, arena, s);
  deg_ = arena.alloc_n<V_ID>(g.nv);
  LUX_OK(hipMemsetAsync(deg_, 0, sizeof(V_ID) * g.nv, s));
  lux_gpu_hist_u32((uint64_t)s, g.ne, g.src, deg_);
  old_ = arena.alloc_n<float>(g.nv);
  new_ = arena.alloc_n<float>(g.nv);
  // init on host: rank/deg (pagerank_gpu.cu:255-259)
  std::vector<V_ID> hdeg(g.nv);
  LUX_OK(hipMemcpyAsync(hdeg.data(), deg_, sizeof(V_ID) * g.nv,
                           hipMemcpyDeviceToHost, s));
  LUX_OK(hipStreamSynchronize(s));
  std::vector<float> hpr(g.nv);
  float rank = 1.0f / g.nv;
  for (V_ID v = 0; v < g.nv; v++)
    hpr[v] = hdeg[v] == 0 ? rank : rank / hdeg[v];
  LUX_OK(hipMemcpyAsync(old_, hpr.data(), sizeof(float) * g.nv,
                           hipMemcpyHostToDevice, s));
  LUX_OK(hipStreamSynchronize(s));
}

void SingleGpuPagerank::iterate(int iters) {
  float init_rank = (1.0f - PR_ALPHA) / g_.nv;
  for (int it = 0; it < iters; it++) {
    LUX_OK(hipMemsetAsync(new_, 0, sizeof(float) * g_.nv, s_));
    if (blocked_.active())
      blocked_.sweep(0, old_, new_, deg_, 0, init_rank, s_);
    else
      lux_gpu_pull_iter((uint64_t)s_, 0, bins_.n0, bins_.bin0, bins_.n1,
                        bins_.bin1, bins_.n2, bins_.bin2, bins_.nbig,
                        bins_.bin2v, row_ptr_, 0, g_.src, old_, new_, deg_,
                        0, init_rank);
    lux_gpu_pull_finish_pr((uint64_t)s_, g_.nv, new_, deg_, 0, init_rank);
    std::swap(old_, new_);
  }
  LUX_OK(hipStreamSynchronize(s_));
}

// ---------------- Push (SSSP / CC) ----------------

SingleGpuPush::SingleGpuPush(const DeviceGraph& g, bool is_min, V_ID source,
                             DeviceArena& arena, hipStream_t s, bool verbose)
    : g_(g), s_(s), is_min_(is_min), verbose_(verbose) {
  row_ptr_ = arena.alloc_n<E_ID>(g.nv + 1);
  lux_gpu_local_row_ptr((uint64_t)s, g.nv, 0, g.col_end, row_ptr_);
  bins_.build(row_ptr_, g.nv, g.ne, arena, s);
  blocked_.build(row_ptr_, g.src, g.nv, g.ne, g.nv, arena, s);
  if (is_min) bits_ = arena.alloc_n<uint32_t>((g.nv + 31) / 32 + 1);
  // push CSR (single partition: transpose over all nv)
  push_row_ptr_ = arena.alloc_n<E_ID>(g.nv + 1);
  push_col_ = arena.alloc_n<V_ID>(g.ne);
  {
    uint32_t* degs;
    E_ID* ends;
    unsigned long long *cursor, *partials;
    LUX_OK(hipMalloc(&degs, sizeof(uint32_t) * g.nv));
    LUX_OK(hipMalloc(&ends, sizeof(E_ID) * g.nv));
    LUX_OK(hipMalloc(&cursor, sizeof(uint64_t) * g.nv));
    LUX_OK(hipMalloc(&partials,
                        sizeof(uint64_t) * lux_gpu_scan_partials_size(g.nv)));
    LUX_OK(hipMemsetAsync(degs, 0, sizeof(uint32_t) * g.nv, s));
    lux_gpu_hist_u32((uint64_t)s, g.ne, g.src, degs);
    lux_gpu_scan_end_offsets((uint64_t)s, g.nv, degs, ends, partials);
    lux_gpu_local_row_ptr((uint64_t)s, g.nv, 0, ends, push_row_ptr_);
    LUX_OK(hipMemcpyAsync(cursor, push_row_ptr_, sizeof(E_ID) * g.nv,
                             hipMemcpyDeviceToDevice, s));
    lux_gpu_csr_scatter((uint64_t)s, g.ne, g.src, row_ptr_, g.nv, 0, cursor,
                        push_col_);
    LUX_OK(hipStreamSynchronize(s));
    hipFree(degs);
    hipFree(ends);
    hipFree(cursor);
    hipFree(partials);
  }
  labels_ = arena.alloc_n<uint32_t>(g.nv);
  snapshot_ = arena.alloc_n<uint32_t>(g.nv);
  uint64_t fq_bytes = frontier_bytes(g.nv);
  fq_ = arena.alloc_n<uint8_t>(fq_bytes);
  new_fq_ = arena.alloc_n<uint8_t>(fq_bytes);
  tmp_fq_ = arena.alloc_n<uint8_t>(fq_bytes);
  // edge-balanced scatter work items (push.hip expand+chunk); bounded by
  // the nv/16 push threshold + per-8192-edge chunk splits
  max_items_ = (uint32_t)(g.nv / SPARSE_THRESHOLD + g.ne / 8192 + 1024);
  items_ = arena.alloc_n<lux_uint2>(max_items_);
  item_counter_ = arena.alloc_n<uint32_t>(4);
  capacity_ = frontier_capacity(g.nv);
  // seed labels + frontier (sssp_gpu.cu:733-744, components_gpu.cu:733-740)
  std::vector<uint32_t> hl(g.nv);
  std::vector<uint8_t> hfq(fq_bytes, 0);
  FrontierHeader* hdr = (FrontierHeader*)hfq.data();
  if (is_min) {
    for (V_ID v = 0; v < g.nv; v++) hl[v] = INF_LABEL;
    hl[source] = 0;
    hdr->type = FrontierHeader::SPARSE_QUEUE;
    hdr->numNodes = 1;
    *(V_ID*)(hfq.data() + sizeof(FrontierHeader)) = source;
  } else {
    for (V_ID v = 0; v < g.nv; v++) hl[v] = v;
    hdr->type = FrontierHeader::DENSE_BITMAP;
    hdr->numNodes = g.nv;
    memset(hfq.data() + sizeof(FrontierHeader), 0xFF, (g.nv + 7) / 8);
  }
  fq_type_ = hdr->type;
  fq_num_ = hdr->numNodes;
  LUX_OK(hipMemcpyAsync(labels_, hl.data(), sizeof(uint32_t) * g.nv,
                           hipMemcpyHostToDevice, s));
  LUX_OK(hipMemcpyAsync(fq_, hfq.data(), fq_bytes, hipMemcpyHostToDevice,
                           s));
  LUX_OK(hipStreamSynchronize(s));
}

V_ID SingleGpuPush::step() {
  LUX_OK(hipMemcpyAsync(snapshot_, labels_, sizeof(uint32_t) * g_.nv,
                           hipMemcpyDeviceToDevice, s_));
  LUX_OK(hipMemsetAsync(new_fq_, 0, sizeof(FrontierHeader), s_));
  bool new_dense = fq_type_ == FrontierHeader::DENSE_BITMAP;
  bool pull_fallback = fq_num_ > g_.nv / SPARSE_THRESHOLD || force_pull_;
  force_pull_ = false;
  auto run_pull = [&]() {
    // dense pull iteration; labels_ serves as both old (all) and new
    // (slice); src-blocked sweeps when the gather window exceeds the LLC
    if (blocked_.active())
      blocked_.sweep(is_min_ ? 1 : 2, snapshot_, labels_, nullptr, 0, 0.0f,
                     s_);
    else
      lux_gpu_pull_iter((uint64_t)s_, is_min_ ? 1 : 2, bins_.n0, bins_.bin0,
                        bins_.n1, bins_.bin1, bins_.n2, bins_.bin2,
                        bins_.nbig, bins_.bin2v, row_ptr_, 0, g_.src,
                        snapshot_, labels_, nullptr, 0, 0.0f);
    bits_stale_ = true;
  };
  if (pull_fallback) {
    new_dense = true;
    run_pull();
  } else {
    LUX_OK(hipMemsetAsync(item_counter_, 0, 16, s_));
    lux_gpu_frontier_expand(
        (uint64_t)s_, fq_type_ == FrontierHeader::DENSE_BITMAP ? 1 : 0, 0,
        fq_type_ == FrontierHeader::DENSE_BITMAP ? g_.nv : fq_num_, fq_,
        nullptr, nullptr, push_row_ptr_, items_, item_counter_, max_items_);
    // second adaptivity axis (python engine parity): the frontier's
    // out-edge volume, counted by the expand kernel (u32 — exact for
    // ne < 2^32), decides push vs a dense pull sweep
    uint32_t hc[4];
    LUX_OK(hipMemcpyAsync(hc, item_counter_, 16, hipMemcpyDeviceToHost,
                          s_));
    LUX_OK(hipStreamSynchronize(s_));
    uint64_t thresh = (is_min_ && bits_) ? g_.ne / 2 : g_.ne / 8;
    if (g_.ne < (1ull << 32) && hc[1] > thresh) {
      pull_fallback = true;
      new_dense = true;
      run_pull();
    } else {
      if ((uint64_t)hc[1] / 16 > capacity_) new_dense = true;
      uint32_t* bits = nullptr;
      if (is_min_ && bits_) {
        if (bits_stale_) {
          lux_gpu_bits_from_labels((uint64_t)s_, g_.nv, labels_, bits_);
          bits_stale_ = false;
        }
        bits = bits_;
      }
      lux_gpu_push_chunk_scatter((uint64_t)s_, is_min_ ? 1 : 0,
                                 new_dense ? 1 : 0, items_, item_counter_,
                                 max_items_, push_row_ptr_, push_col_,
                                 snapshot_, snapshot_, labels_, 0, new_fq_,
                                 capacity_, bits);
    }
  }
  FrontierHeader hh;
  if (new_dense) {
    lux_gpu_build_bitmap((uint64_t)s_, g_.nv, snapshot_, labels_, new_fq_);
    LUX_OK(hipMemcpyAsync(&hh, new_fq_, 8, hipMemcpyDeviceToHost, s_));
    LUX_OK(hipStreamSynchronize(s_));
    if (hh.numNodes < capacity_) {
      LUX_OK(hipMemcpyAsync(tmp_fq_, new_fq_, frontier_bytes(g_.nv),
                               hipMemcpyDeviceToDevice, s_));
      LUX_OK(hipMemsetAsync(new_fq_, 0, sizeof(FrontierHeader), s_));
      lux_gpu_d2s((uint64_t)s_, g_.nv, 0, tmp_fq_, new_fq_);
      LUX_OK(hipMemcpyAsync(&hh, new_fq_, 8, hipMemcpyDeviceToHost, s_));
      LUX_OK(hipStreamSynchronize(s_));
      new_dense = false;
    }
  } else {
    LUX_OK(hipMemcpyAsync(&hh, new_fq_, 8, hipMemcpyDeviceToHost, s_));
    LUX_OK(hipStreamSynchronize(s_));
    if (hh.numNodes >= capacity_) {
      new_dense = true;
      LUX_OK(hipMemsetAsync(new_fq_, 0, sizeof(FrontierHeader), s_));
      lux_gpu_build_bitmap((uint64_t)s_, g_.nv, snapshot_, labels_, new_fq_);
      LUX_OK(hipMemcpyAsync(&hh, new_fq_, 8, hipMemcpyDeviceToHost, s_));
      LUX_OK(hipStreamSynchronize(s_));
    }
  }
  if (!pull_fallback) {
    // expand overflow guard: truncated work items
    // mean unrelaxed edges — recover loudly with a forced pull iteration
    // (re-relaxes every edge; labels are monotone so this is safe)
    uint32_t hc[4];
    LUX_OK(hipMemcpyAsync(hc, item_counter_, 16, hipMemcpyDeviceToHost, s_));
    LUX_OK(hipStreamSynchronize(s_));
    if (hc[3] || hc[0] > max_items_) {
      fprintf(stderr,
              "[lux] frontier expand overflow (%u items > cap %u): forcing "
              "a pull iteration to recover\n", hc[0], max_items_);
      force_pull_ = true;
    }
  }
  fq_type_ = new_dense ? FrontierHeader::DENSE_BITMAP
                       : FrontierHeader::SPARSE_QUEUE;
  fq_num_ = hh.numNodes;
  std::swap(fq_, new_fq_);
  if (verbose_)
    printf("iter %d: activeNodes(%u) %s\n", iters_, fq_num_,
           new_dense ? "dense" : "sparse");
  return fq_num_;
}

int SingleGpuPush::run(int max_iters) {
  while (true) {
    V_ID n = step();
    iters_++;
    if (n == 0 && !force_pull_) break;  // overflow: don't terminate early
    if (max_iters && iters_ >= max_iters) break;
  }
  return iters_;
}

uint64_t SingleGpuPush::check() {
  unsigned long long* mistakes;
  LUX_OK(hipMalloc(&mistakes, 8));
  LUX_OK(hipMemsetAsync(mistakes, 0, 8, s_));
  lux_gpu_check((uint64_t)s_, is_min_ ? 1 : 0, g_.nv, 0, row_ptr_, g_.src,
                labels_, mistakes);
  unsigned long long h;
  LUX_OK(hipMemcpyAsync(&h, mistakes, 8, hipMemcpyDeviceToHost, s_));
  LUX_OK(hipStreamSynchronize(s_));
  hipFree(mistakes);
  return h;
}

// ---------------- CC (union-find) ----------------

SingleGpuCCUnionFind::SingleGpuCCUnionFind(const DeviceGraph& g,
                                           DeviceArena& arena,
                                           hipStream_t s)
    : g_(g), s_(s) {
  row_ptr_ = arena.alloc_n<E_ID>(g.nv + 1);
  lux_gpu_local_row_ptr((uint64_t)s, g.nv, 0, g.col_end, row_ptr_);
  bins_.build(row_ptr_, g.nv, g.ne, arena, s);
  parent_ = arena.alloc_n<V_ID>(g.nv);
  labels_ = arena.alloc_n<V_ID>(g.nv);
  gbits_ = arena.alloc_n<uint32_t>((g.nv + 31) / 32);
}

void SingleGpuCCUnionFind::run() {
  // parent = iota
  std::vector<V_ID> h(g_.nv);
  for (V_ID v = 0; v < g_.nv; v++) h[v] = v;
  LUX_OK(hipMemcpyAsync(parent_, h.data(), sizeof(V_ID) * g_.nv,
                        hipMemcpyHostToDevice, s_));
  for (uint32_t k = 0; k < 2; k++)  // Afforest sample-hook rounds
    lux_gpu_uf_union_kth((uint64_t)s_, g_.nv, row_ptr_, g_.src, 0, parent_,
                         k);
  lux_gpu_uf_flatten((uint64_t)s_, g_.nv, parent_, labels_);
  // giant root by host sampling
  V_ID nsamp = g_.nv < 4096 ? g_.nv : 4096;
  V_ID stride = g_.nv / nsamp;
  std::vector<V_ID> samp(nsamp);
  LUX_OK(hipMemcpy2DAsync(samp.data(), sizeof(V_ID), labels_,
                          (size_t)stride * sizeof(V_ID), sizeof(V_ID),
                          nsamp, hipMemcpyDeviceToHost, s_));
  LUX_OK(hipStreamSynchronize(s_));
  std::sort(samp.begin(), samp.end());
  V_ID giant = samp[0], best = 1, run = 1;
  for (size_t i = 1; i < samp.size(); i++) {
    run = samp[i] == samp[i - 1] ? run + 1 : 1;
    if (run > best) { best = run; giant = samp[i]; }
  }
  lux_gpu_cc_giant_bits((uint64_t)s_, g_.nv, labels_, giant, gbits_);
  lux_gpu_uf_union_binned((uint64_t)s_, bins_.n0, bins_.bin0, bins_.n1,
                          bins_.bin1, bins_.n2, bins_.bin2, row_ptr_,
                          g_.src, 0, parent_, gbits_);
  lux_gpu_uf_flatten((uint64_t)s_, g_.nv, parent_, labels_);
  LUX_OK(hipStreamSynchronize(s_));
}

uint64_t SingleGpuCCUnionFind::check() {
  unsigned long long* mistakes;
  LUX_OK(hipMalloc(&mistakes, 8));
  LUX_OK(hipMemsetAsync(mistakes, 0, 8, s_));
  lux_gpu_check((uint64_t)s_, 0, g_.nv, 0, row_ptr_, g_.src,
                (const uint32_t*)labels_, mistakes);
  unsigned long long h = 0;
  LUX_OK(hipMemcpyAsync(&h, mistakes, 8, hipMemcpyDeviceToHost, s_));
  LUX_OK(hipStreamSynchronize(s_));
  hipFree(mistakes);
  return h;
}

// ---------------- CF ----------------

// Stable-partition the bin lists at local row boundary lb (rows < lb
// first). Host roundtrip; init-time only.
BinSplit split_bins_at(Bins& b, V_ID lb, hipStream_t s) {
  LUX_OK(hipStreamSynchronize(s));
  BinSplit out;
  auto part_v = [&](V_ID* dev, uint32_t n, uint32_t* cnt) {
    if (!n) return;
    std::vector<V_ID> h(n), lo, hi;
    LUX_OK(hipMemcpy(h.data(), dev, sizeof(V_ID) * n,
                     hipMemcpyDeviceToHost));
    for (V_ID v : h) (v < lb ? lo : hi).push_back(v);
    *cnt = (uint32_t)lo.size();
    lo.insert(lo.end(), hi.begin(), hi.end());
    LUX_OK(hipMemcpy(dev, lo.data(), sizeof(V_ID) * n,
                     hipMemcpyHostToDevice));
  };
  part_v(b.bin0, b.n0, &out.n0u);
  part_v(b.bin1, b.n1, &out.n1u);
  part_v(b.bin2v, b.nbig, &out.nbigu);
  if (b.n2) {
    std::vector<lux_uint2> h(b.n2), lo, hi;
    LUX_OK(hipMemcpy(h.data(), b.bin2, sizeof(lux_uint2) * b.n2,
                     hipMemcpyDeviceToHost));
    for (auto& e : h) (e.x < lb ? lo : hi).push_back(e);
    out.n2u = (uint32_t)lo.size();
    lo.insert(lo.end(), hi.begin(), hi.end());
    LUX_OK(hipMemcpy(b.bin2, lo.data(), sizeof(lux_uint2) * b.n2,
                     hipMemcpyHostToDevice));
  }
  return out;
}

SingleGpuCF::SingleGpuCF(const DeviceGraph& g, int K, DeviceArena& arena,
                         hipStream_t s, bool als, V_ID n_users)
    : g_(g), s_(s), K_(K), als_(als), nu_(als ? n_users : 0) {
  row_ptr_ = arena.alloc_n<E_ID>(g.nv + 1);
  lux_gpu_local_row_ptr((uint64_t)s, g.nv, 0, g.col_end, row_ptr_);
  bins_.build(row_ptr_, g.nv, g.ne, arena, s);
  if (nu_) split_ = split_bins_at(bins_, nu_ < g.nv ? nu_ : g.nv, s);
  old_ = arena.alloc_n<float>((size_t)g.nv * K);
  new_ = arena.alloc_n<float>((size_t)g.nv * K);
  std::vector<float> h((size_t)g.nv * K, sqrtf(1.0f / K));
  if (als_)  // jittered init (see als_init_val); SGD keeps the constant
    for (size_t i = 0; i < h.size(); i++) h[i] = als_init_val(i, K);
  LUX_OK(hipMemcpyAsync(old_, h.data(), sizeof(float) * h.size(),
                           hipMemcpyHostToDevice, s));
  if (als_ && bins_.nbig) {
    // hub scratch: slot map (local id -> scratch slot), Gram + rhs.
    // With alternation, slots are PHASE-relative: the item half-sweep
    // passes gram/rhs offset by nbigu, so item hub i maps to i - nbigu.
    hubidx_ = arena.alloc_n<int>(g.nv);
    std::vector<V_ID> hubs(bins_.nbig);
    LUX_OK(hipMemcpy(hubs.data(), bins_.bin2v,
                     sizeof(V_ID) * bins_.nbig, hipMemcpyDeviceToHost));
    std::vector<int> hidx(g.nv, -1);
    for (uint32_t i = 0; i < bins_.nbig; i++)
      hidx[hubs[i]] = (int)(nu_ && i >= split_.nbigu ? i - split_.nbigu : i);
    LUX_OK(hipMemcpyAsync(hubidx_, hidx.data(), sizeof(int) * g.nv,
                          hipMemcpyHostToDevice, s));
    LUX_OK(hipStreamSynchronize(s));  // hidx leaves scope
    gram_ = arena.alloc_n<float>((size_t)bins_.nbig * 64 * 64);
    rhs_ = arena.alloc_n<float>((size_t)bins_.nbig * 64);
  }
  LUX_OK(hipStreamSynchronize(s));
}

void SingleGpuCF::iterate(int iters) {
  size_t n = (size_t)g_.nv * K_;
  for (int it = 0; it < iters; it++) {
    if (als_) {
      // seed: solved rows are overwritten, deg-0 rows keep their vector
      LUX_OK(hipMemcpyAsync(new_, old_, sizeof(float) * n,
                            hipMemcpyDeviceToDevice, s_));
      if (bins_.nbig) {
        LUX_OK(hipMemsetAsync(gram_, 0,
                              sizeof(float) * (size_t)bins_.nbig * 64 * 64,
                              s_));
        LUX_OK(hipMemsetAsync(rhs_, 0,
                              sizeof(float) * (size_t)bins_.nbig * 64, s_));
      }
      if (nu_) {
        // Gauss-Seidel alternation: users from old_, then items from the
        // UPDATED users already in new_ (item rows of new_ still hold the
        // seeded old values, which no item row reads)
        lux_gpu_cf_als_iter((uint64_t)s_, split_.n0u, bins_.bin0,
                            split_.n1u, bins_.bin1, split_.n2u, bins_.bin2,
                            split_.nbigu, bins_.bin2v, hubidx_, gram_, rhs_,
                            row_ptr_, g_.src, g_.weight, old_, nullptr,
                            new_, 0, K_);
        lux_gpu_cf_als_iter(
            (uint64_t)s_, bins_.n0 - split_.n0u, bins_.bin0 + split_.n0u,
            bins_.n1 - split_.n1u, bins_.bin1 + split_.n1u,
            bins_.n2 - split_.n2u, bins_.bin2 + split_.n2u,
            bins_.nbig - split_.nbigu, bins_.bin2v + split_.nbigu, hubidx_,
            gram_ ? gram_ + (size_t)split_.nbigu * 64 * 64 : nullptr,
            rhs_ ? rhs_ + (size_t)split_.nbigu * 64 : nullptr, row_ptr_,
            g_.src, g_.weight, new_, nullptr, new_, 0, K_);
      } else {
        lux_gpu_cf_als_iter((uint64_t)s_, bins_.n0, bins_.bin0, bins_.n1,
                            bins_.bin1, bins_.n2, bins_.bin2, bins_.nbig,
                            bins_.bin2v, hubidx_, gram_, rhs_, row_ptr_,
                            g_.src, g_.weight, old_, nullptr, new_, 0, K_);
      }
    } else {
      // SGD contract (cf.hip): output pre-seeded old*(1-GAMMA*LAMBDA),
      // sweeps add GAMMA*acc
      lux_gpu_cf_seed((uint64_t)s_, n, old_, new_);
      lux_gpu_cf_iter((uint64_t)s_, bins_.n0, bins_.bin0, bins_.n1,
                      bins_.bin1, bins_.n2, bins_.bin2, bins_.nbig,
                      bins_.bin2v, row_ptr_, g_.src, g_.weight, old_, new_,
                      0, K_);
    }
    std::swap(old_, new_);
  }
  LUX_OK(hipStreamSynchronize(s_));
}

}  // namespace lux
