// Pull-model engine: degree-binned CSC gather kernels for gfx950.
//
// Replaces the reference's one-size block-scan CSC walk (pr_kernel,
// pagerank_gpu.cu:49-102; sssp_pull_kernel, sssp_gpu.cu:85-130) with a
// three-bin design shaped for CDNA4 and power-law graphs:
//   bin0 (deg <  T1): one thread per dst vertex, grid-stride.
//   bin1 (T1..T2):    one 64-lane wave per dst vertex; lanes stride the
//                     contiguous in-edge range (coalesced 256 B/instr col
//                     reads), shuffle-reduce, lane 0 writes.
//   bin2 (deg >= T2): hub vertices split into CHUNK_EDGES-edge chunks, one
//                     256-thread block per chunk, block-reduce + one global
//                     atomic per block (guideline: reduce first, atomic once).
// Bins are built once at init (degrees are static); per iteration we launch
// prep (bin2 only) + 3 bin kernels on one stream. The same machinery runs
// PageRank's float sum and SSSP/CC's u32 min/max dense fallback.
#include "gpu_common.h"

namespace lux {

constexpr V_ID T1 = 32;
constexpr V_ID T2 = 2048;
constexpr V_ID CHUNK_EDGES = 8192;

enum PullMode { PR_SUM = 0, LAB_MIN = 1, LAB_MAX = 2, CF_SGD = 3 };

// ---------------- bin building ----------------

// LDS-aggregated bin build: per-tile LDS counters, ONE global atomic per
// bin per tile (global-atomic contention on 3 words made the naive version
// ~250x slower at nv=2^27 — guideline 12: reduce first, atomic once).
__global__ void build_bins_kernel(V_ID vp, const E_ID* row_ptr,
                                  V_ID* bin0, V_ID* bin1, uint2* bin2,
                                  V_ID* bin2v, uint32_t* counters) {
  __shared__ uint32_t lcnt[4], lbase[4];
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t tile = (uint64_t)blockIdx.x * blockDim.x; tile < vp;
       tile += stride) {
    uint64_t v = tile + threadIdx.x;
    if (threadIdx.x < 4) lcnt[threadIdx.x] = 0;
    __syncthreads();
    int my_bin = -1;
    uint32_t lpos = 0, nchunks = 0, lpos_c = 0;
    if (v < vp) {
      E_ID deg = row_ptr[v + 1] - row_ptr[v];
      if (deg == 0) {
        // not binned: rows with no edges in this (block-)CSC are covered by
        // the seed/finish pass, so sweeps touch only active rows
      } else if (deg < T1) {
        my_bin = 0;
        lpos = atomicAdd(&lcnt[0], 1u);
      } else if (deg < T2) {
        my_bin = 1;
        lpos = atomicAdd(&lcnt[1], 1u);
      } else {
        my_bin = 3;
        lpos = atomicAdd(&lcnt[3], 1u);
        nchunks = (uint32_t)((deg + CHUNK_EDGES - 1) / CHUNK_EDGES);
        lpos_c = atomicAdd(&lcnt[2], nchunks);
      }
    }
    __syncthreads();
    if (threadIdx.x < 4)
      lbase[threadIdx.x] = lcnt[threadIdx.x]
                               ? atomicAdd(&counters[threadIdx.x],
                                           lcnt[threadIdx.x])
                               : 0;
    __syncthreads();
    if (my_bin == 0) bin0[lbase[0] + lpos] = (V_ID)v;
    else if (my_bin == 1) bin1[lbase[1] + lpos] = (V_ID)v;
    else if (my_bin == 3) {
      bin2v[lbase[3] + lpos] = (V_ID)v;
      uint32_t base = lbase[2] + lpos_c;
      for (uint32_t c = 0; c < nchunks; c++)
        bin2[base + c] = make_uint2((V_ID)v, c);
    }
    __syncthreads();
  }
}

// ---------------- per-mode value semantics ----------------

template <PullMode M> struct Val;
template <> struct Val<PR_SUM> {
  using T = float;
  static constexpr bool SKIP_SETTLED = false;
  static __device__ __forceinline__ T ident() { return 0.0f; }
  static __device__ __forceinline__ T map(T s) { return s; }
  static __device__ __forceinline__ T comb(T a, T b) { return a + b; }
  static __device__ __forceinline__ void atom(T* p, T v) { atomicAdd(p, v); }
  static __device__ __forceinline__ T reduce_wave(T v) {
    return wave_reduce_sum(v);
  }
};
template <> struct Val<LAB_MIN> {
  using T = uint32_t;
  // hop-SSSP (BFS) invariant: a finite label is the true depth and can
  // never improve under synchronized iterations — settled rows skip their
  // whole edge range (newv keeps the seeded own label). Only valid for the
  // +1 hop metric; LAB_MAX (CC) labels keep moving and never skip.
  static constexpr bool SKIP_SETTLED = true;
  static __device__ __forceinline__ T ident() { return INF_LABEL; }
  static __device__ __forceinline__ T map(T s) {
    return s == INF_LABEL ? INF_LABEL : s + 1;  // hop relaxation (+1)
  }
  static __device__ __forceinline__ T comb(T a, T b) { return a < b ? a : b; }
  static __device__ __forceinline__ void atom(T* p, T v) { atomicMin(p, v); }
  static __device__ __forceinline__ T reduce_wave(T v) {
    return wave_reduce_min(v);
  }
};
template <> struct Val<LAB_MAX> {
  using T = uint32_t;
  static constexpr bool SKIP_SETTLED = false;
  static __device__ __forceinline__ T ident() { return 0; }
  static __device__ __forceinline__ T map(T s) { return s; }
  static __device__ __forceinline__ T comb(T a, T b) { return a > b ? a : b; }
  static __device__ __forceinline__ void atom(T* p, T v) { atomicMax(p, v); }
  static __device__ __forceinline__ T reduce_wave(T v) {
    return wave_reduce_max(v);
  }
};

// Epilogue: PR computes (1-a)/nv + a*sum then stores /out_degree
// (pagerank_gpu.cu:97-100); labels fold the dst's own old label in.
template <PullMode M>
__device__ __forceinline__ typename Val<M>::T finish(
    typename Val<M>::T acc, typename Val<M>::T own, float init_rank,
    V_ID deg) {
  if (M == PR_SUM) {
    float pr = init_rank + PR_ALPHA * (float)acc;
    return deg != 0 ? pr / (float)deg : pr;
  }
  return Val<M>::comb(acc, own);
}

struct PullArgs {
  const void* row_ptr;   // RowT[vp+1], local 0-based: u64, or u32 for the
                         // blocked path (block-local offsets < 2^32 — half
                         // the per-row sweep traffic, NOTES_r2 item 1)
  const V_ID* col;       // u32[ep], local
  const void* oldv;      // full nv
  void* newv;            // vp
  const V_ID* deg;       // u32[nv] out-degrees (PR) or null
  V_ID row_left;
  float init_rank;
};

// Result store for the non-atomic bins. The iteration contract: newv is
// pre-seeded (PR: zeros; labels: the old label slice) before the sweep(s),
// every sweep — blocked or not — folds its partial in, and PR's epilogue
// runs once at the end (pull_finish_kernel). One writer per row per sweep,
// so the fold needs no atomics.
template <PullMode M>
__device__ __forceinline__ void store_result(typename Val<M>::T* slot,
                                             typename Val<M>::T acc) {
  *slot = Val<M>::comb(acc, *slot);
}

// Edge-range accumulation with a 4-deep software pipeline: the rolled
// col[j] -> oldv[col[j]] chain is two dependent loads with ONE iteration
// in flight (the dynamic trip count stops the compiler from unrolling);
// batching 4 col reads then 4 independent gathers quadruples the memory
// parallelism per lane (same fix as cf.hip cf_stage_tile).
template <PullMode M>
__device__ __forceinline__ typename Val<M>::T gather_range(
    const typename Val<M>::T* oldv, const V_ID* col, E_ID j, E_ID e,
    E_ID step) {
  using V = Val<M>;
  using T = typename V::T;
  T a0 = V::ident(), a1 = V::ident(), a2 = V::ident(), a3 = V::ident();
  for (; j + 3 * step < e; j += 4 * step) {
    V_ID c0 = col[j], c1 = col[j + step], c2 = col[j + 2 * step],
         c3 = col[j + 3 * step];
    a0 = V::comb(a0, V::map(oldv[c0]));
    a1 = V::comb(a1, V::map(oldv[c1]));
    a2 = V::comb(a2, V::map(oldv[c2]));
    a3 = V::comb(a3, V::map(oldv[c3]));
  }
  for (; j < e; j += step) a0 = V::comb(a0, V::map(oldv[col[j]]));
  return V::comb(V::comb(a0, a1), V::comb(a2, a3));
}

// ---- bin0: thread per vertex ----
template <PullMode M, typename RowT>
__global__ void pull_thread_kernel(uint32_t n0, const V_ID* bin0,
                                   PullArgs a) {
  using V = Val<M>;
  using T = typename V::T;
  const T* oldv = (const T*)a.oldv;
  T* newv = (T*)a.newv;
  const RowT* row_ptr = (const RowT*)a.row_ptr;
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < n0;
       i += stride) {
    V_ID v = bin0[i];
    if (V::SKIP_SETTLED && oldv[a.row_left + v] != (T)INF_LABEL)
      continue;  // settled hop label: newv keeps the seed
    E_ID b = row_ptr[v], e = row_ptr[v + 1];
    T acc = gather_range<M>(oldv, a.col, b, e, 1);
    store_result<M>(&newv[v], acc);
  }
}

// ---- bin1: wave per vertex ----
template <PullMode M, typename RowT>
__global__ void pull_wave_kernel(uint32_t n1, const V_ID* bin1, PullArgs a) {
  using V = Val<M>;
  using T = typename V::T;
  const T* oldv = (const T*)a.oldv;
  T* newv = (T*)a.newv;
  const RowT* row_ptr = (const RowT*)a.row_ptr;
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t wave_id = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint64_t nwaves = ((uint64_t)gridDim.x * blockDim.x) / WAVE;
  for (uint64_t i = wave_id; i < n1; i += nwaves) {
    V_ID v = bin1[i];
    if (V::SKIP_SETTLED && oldv[a.row_left + v] != (T)INF_LABEL)
      continue;
    E_ID b = row_ptr[v], e = row_ptr[v + 1];
    T acc = gather_range<M>(oldv, a.col, b + lane, e, WAVE);
    acc = V::reduce_wave(acc);
    if (lane == 0)
      store_result<M>(&newv[v], acc);
  }
}

// ---- bin2: chunk accumulation (hubs) ----
template <PullMode M, typename RowT>
__global__ void pull_chunk_kernel(uint32_t n2, const uint2* bin2,
                                  PullArgs a) {
  using V = Val<M>;
  using T = typename V::T;
  __shared__ T lds[BLOCK / WAVE];
  const T* oldv = (const T*)a.oldv;
  T* newv = (T*)a.newv;
  const RowT* row_ptr = (const RowT*)a.row_ptr;
  for (uint32_t i = blockIdx.x; i < n2; i += gridDim.x) {
    uint2 ent = bin2[i];
    V_ID v = ent.x;
    if (V::SKIP_SETTLED && oldv[a.row_left + v] != (T)INF_LABEL)
      continue;
    E_ID b = (E_ID)row_ptr[v] + (E_ID)ent.y * CHUNK_EDGES;
    E_ID e = row_ptr[v + 1];
    if (e > b + CHUNK_EDGES) e = b + CHUNK_EDGES;
    T acc = gather_range<M>(oldv, a.col, b + threadIdx.x, e, blockDim.x);
    // block reduce (sum mode) or wave+lds fold (min/max)
    int lane = threadIdx.x & (WAVE - 1);
    int wid = threadIdx.x >> 6;
    acc = V::reduce_wave(acc);
    if (lane == 0) lds[wid] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      acc = lds[0];
      for (int w = 1; w < (int)(blockDim.x / WAVE); w++)
        acc = V::comb(acc, lds[w]);
      V::atom(&newv[v], acc);
    }
    __syncthreads();
  }
}


template <PullMode M, typename RowT>
static void pull_iter(hipStream_t s, uint32_t n0, const V_ID* bin0,
                      uint32_t n1, const V_ID* bin1, uint32_t n2,
                      const uint2* bin2, uint32_t nbig, const V_ID* bin2v,
                      const PullArgs& a) {
  if (nbig) {
    hipLaunchKernelGGL((pull_chunk_kernel<M, RowT>),
                       dim3(n2 > MAX_GRID ? MAX_GRID : n2), dim3(BLOCK), 0, s,
                       n2, bin2, a);
  }
  if (n1)
    hipLaunchKernelGGL((pull_wave_kernel<M, RowT>),
                       dim3(grid_for((uint64_t)n1 * WAVE)), dim3(BLOCK), 0, s,
                       n1, bin1, a);
  if (n0)
    hipLaunchKernelGGL((pull_thread_kernel<M, RowT>), dim3(grid_for(n0)),
                       dim3(BLOCK), 0, s, n0, bin0, a);
}

// PR epilogue over the whole partition: pr = (1-a)/nv + a*sum, stored
// divided by out-degree (pagerank_gpu.cu:97-100, :255-259). Covers rows
// with no in-edges too (sum stays 0 from the seed).
__global__ void pull_finish_pr_kernel(V_ID vp, float* newv,
                                      const V_ID* deg, V_ID row_left,
                                      float init_rank) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t v = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; v < vp;
       v += stride) {
    float pr = init_rank + PR_ALPHA * newv[v];
    V_ID d = deg[row_left + v];
    newv[v] = d != 0 ? pr / (float)d : pr;
  }
}

}  // namespace lux

// ---------------- C ABI ----------------

using namespace lux;

extern "C" {

void lux_gpu_build_bins(uint64_t stream, uint32_t vp, const E_ID* row_ptr,
                        V_ID* bin0, V_ID* bin1, uint2* bin2, V_ID* bin2v,
                        uint32_t* counters /*pre-zeroed u32[4]*/) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(build_bins_kernel, dim3(grid_for(vp)), dim3(BLOCK), 0, s,
                     vp, row_ptr, bin0, bin1, bin2, bin2v, counters);
  LUX_POST_LAUNCH(stream);
}

// mode: 0 = PageRank float-sum, 1 = u32 min (SSSP dense), 2 = u32 max (CC).
// row_u32: row_ptr is u32[vp+1] (block-local offsets) instead of u64.
void lux_gpu_pull_iter(uint64_t stream, int mode, uint32_t n0,
                       const V_ID* bin0, uint32_t n1, const V_ID* bin1,
                       uint32_t n2, const uint2* bin2, uint32_t nbig,
                       const V_ID* bin2v, const void* row_ptr, int row_u32,
                       const V_ID* col, const void* oldv, void* newv,
                       const V_ID* deg, V_ID row_left, float init_rank) {
  hipStream_t s = (hipStream_t)stream;
  PullArgs a{row_ptr, col, oldv, newv, deg, row_left, init_rank};
#define LUX_PI(M_)                                                            do {                                                                          if (row_u32)                                                                  pull_iter<M_, uint32_t>(s, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v,                               a);                                               else                                                                          pull_iter<M_, uint64_t>(s, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v,                               a);                                             } while (0)
  switch (mode) {
    case 0: LUX_PI(PR_SUM); break;
    case 1: LUX_PI(LAB_MIN); break;
    case 2: LUX_PI(LAB_MAX); break;
  }
#undef LUX_PI
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_pull_finish_pr(uint64_t stream, V_ID vp, float* newv,
                            const V_ID* deg, V_ID row_left,
                            float init_rank) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(pull_finish_pr_kernel, dim3(grid_for(vp)), dim3(BLOCK),
                     0, s, vp, newv, deg, row_left, init_rank);
  LUX_POST_LAUNCH(stream);
}

}  // extern "C"
