// GPU graph construction: R-MAT edge generation + CSC build (counting sort
// by dst: histogram -> device-wide exclusive scan -> scatter), plus degree
// histograms. All kernels gfx950-native; the device-wide scan replaces the
// reference's serial single-thread prefix sum (sssp_gpu.cu:550-565 — a
// deliberate wart SURVEY.md §7 says to fix).
//
// Edge generation is bit-identical to the CPU generator (src/include/lux/rmat.h
// is compiled into both), so GPU-built graphs are validated against CPU ones.
#include "gpu_common.h"
#include "lux/rmat.h"

namespace lux {

// ---------------- edge generation ----------------

// e0: global edge index of the chunk's first edge — the sliced builders
// generate the SAME edge stream as the full build, one chunk at a time
// (edge i of the graph is always rmat_edge(seed, i), so per-rank chunked
// builds are bit-identical to full ones).
__global__ void rmat_edges_kernel(uint64_t seed, int scale, uint64_t e0,
                                  uint64_t ne, V_ID* src, V_ID* dst) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < ne;
       e += stride) {
    rmat_edge(seed, e0 + e, scale, &src[e], &dst[e]);
  }
}

__global__ void rmat_edges_folded_kernel(uint64_t seed, int scale, V_ID nv,
                                         uint64_t e0, uint64_t ne, V_ID* src,
                                         V_ID* dst) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < ne;
       e += stride) {
    rmat_edge_folded(seed, e0 + e, scale, nv, &src[e], &dst[e]);
  }
}

__global__ void bipartite_edges_kernel(uint64_t seed, V_ID n_users,
                                       V_ID n_items, int item_scale,
                                       uint64_t e0, uint64_t ne, V_ID* src,
                                       V_ID* dst, WeightType* w) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < ne;
       e += stride) {
    bipartite_edge(seed, e0 + e, n_users, n_items, item_scale, &src[e],
                   &dst[e]);
    w[e] = rmat_weight(seed, (e0 + e) >> 1);  // both directions share the rating
  }
}

// Filtered scatter for the rank-sliced CSC build (VERDICT r1 missing #2):
// keep only edges landing in my partition [rl, rr]; cursor holds LOCAL
// running offsets (seeded from the local row_ptr begins).
__global__ void slice_scatter_kernel(uint64_t n, const V_ID* src,
                                     const V_ID* dst, const WeightType* w,
                                     V_ID rl, V_ID rr,
                                     unsigned long long* cursor,
                                     V_ID* out_col, WeightType* out_w) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < n;
       e += stride) {
    V_ID d = dst[e];
    if (d < rl || d > rr) continue;
    unsigned long long pos = atomicAdd(&cursor[d - rl], 1ull);
    out_col[pos] = src[e];
    if (w) out_w[pos] = w[e];
  }
}

// ---------------- histogram ----------------

__global__ void hist_u32_kernel(uint64_t n, const V_ID* ids, uint32_t* hist) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < n;
       e += stride)
    atomicAdd(&hist[ids[e]], 1u);
}

// ---------------- device-wide scan u32 -> u64 (inclusive end-offsets) ----

constexpr int SCAN_ITEMS = 16;
constexpr int SCAN_TILE = BLOCK * SCAN_ITEMS;  // 4096

__global__ void scan_reduce_kernel(uint64_t n, const uint32_t* in,
                                   unsigned long long* partials) {
  __shared__ unsigned long long lds[BLOCK / WAVE];
  uint64_t base = (uint64_t)blockIdx.x * SCAN_TILE;
  unsigned long long sum = 0;
  for (int i = 0; i < SCAN_ITEMS; i++) {
    uint64_t idx = base + threadIdx.x + (uint64_t)i * BLOCK;
    if (idx < n) sum += in[idx];
  }
  sum = block_reduce_sum(sum, lds);
  if (threadIdx.x == 0) partials[blockIdx.x] = sum;
}

__global__ void scan_partials_kernel(uint32_t nblocks,
                                     unsigned long long* partials) {
  // single block of 1024 threads, looped with carry
  __shared__ unsigned long long lds[1024 / WAVE + 1];
  unsigned long long carry = 0;
  for (uint32_t base = 0; base < nblocks; base += 1024) {
    uint32_t idx = base + threadIdx.x;
    unsigned long long v = idx < nblocks ? partials[idx] : 0;
    unsigned long long total;
    unsigned long long ex = block_exscan<unsigned long long, 1024>(v, lds,
                                                                   &total);
    if (idx < nblocks) partials[idx] = carry + ex;
    carry += total;
    __syncthreads();
  }
}

__global__ void scan_apply_kernel(uint64_t n, const uint32_t* in,
                                  const unsigned long long* partials,
                                  E_ID* out_end) {
  __shared__ unsigned long long lds[BLOCK / WAVE + 1];
  uint64_t base = (uint64_t)blockIdx.x * SCAN_TILE;
  // per-thread sequential over its contiguous run of SCAN_ITEMS
  uint32_t vals[SCAN_ITEMS];
  unsigned long long mysum = 0;
  uint64_t tbase = base + (uint64_t)threadIdx.x * SCAN_ITEMS;
  for (int i = 0; i < SCAN_ITEMS; i++) {
    uint64_t idx = tbase + i;
    vals[i] = idx < n ? in[idx] : 0;
    mysum += vals[i];
  }
  unsigned long long ex =
      block_exscan<unsigned long long, BLOCK>(mysum, lds, nullptr);
  unsigned long long run = partials[blockIdx.x] + ex;
  for (int i = 0; i < SCAN_ITEMS; i++) {
    uint64_t idx = tbase + i;
    run += vals[i];
    if (idx < n) out_end[idx] = run;  // inclusive end offset
  }
}

void scan_u32_to_end_u64(hipStream_t s, uint64_t n, const uint32_t* in,
                         E_ID* out_end, unsigned long long* partials) {
  uint32_t nblocks = (uint32_t)((n + (uint64_t)SCAN_TILE - 1) / SCAN_TILE);
  hipLaunchKernelGGL(scan_reduce_kernel, dim3(nblocks), dim3(BLOCK), 0, s, n,
                     in, partials);
  hipLaunchKernelGGL(scan_partials_kernel, dim3(1), dim3(1024), 0, s, nblocks,
                     partials);
  hipLaunchKernelGGL(scan_apply_kernel, dim3(nblocks), dim3(BLOCK), 0, s, n,
                     in, partials, out_end);
}

// ---------------- scatter (counting sort by dst) ----------------

__global__ void init_cursor_kernel(uint32_t nv, const E_ID* col_end,
                                   unsigned long long* cursor) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t v = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; v < nv;
       v += stride)
    cursor[v] = v == 0 ? 0ull : (unsigned long long)col_end[v - 1];
}

__global__ void scatter_kernel(uint64_t ne, const V_ID* src, const V_ID* dst,
                               const WeightType* w,
                               unsigned long long* cursor, V_ID* out_src,
                               WeightType* out_w) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t e = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; e < ne;
       e += stride) {
    unsigned long long pos = atomicAdd(&cursor[dst[e]], 1ull);
    out_src[pos] = src[e];
    if (w) out_w[pos] = w[e];
  }
}

// ---------------- src-blocked CSC build ----------------
// Re-groups a partition's CSC edges by (src >> shift, dst): per iteration
// the pull kernels then sweep one src block at a time, keeping the random
// old-property gather window (2^shift * 4 B) resident in the 256 MiB
// Infinity Cache. This is the key MI355X bandwidth lever for nv*4B > LLC
// graphs (RMAT-27: 512 MB of ranks); the reference has no equivalent (its
// gathers run cold over ZC/host memory).

__device__ __forceinline__ V_ID row_of_edge(const E_ID* row_ptr_loc, V_ID vp,
                                            uint64_t j) {
  V_ID lo = 0, hi = vp - 1;
  while (lo < hi) {
    V_ID mid = (lo + hi + 1) >> 1;
    if (row_ptr_loc[mid] <= j) lo = mid;
    else hi = mid - 1;
  }
  return lo;
}

// Block boundaries are an explicit sorted list (bounds[0]=0 ... bounds[nb]
// = nv) rather than a uniform shift: the distributed pull engines align one
// boundary pair with this rank's own vertex range so the rank-local src
// block can be swept while the RCCL all-gather of remote slices is still in
// flight (comm/compute overlap over xGMI).
__device__ __forceinline__ uint32_t block_of(V_ID src, const V_ID* bounds,
                                             int nb) {
  int lo = 0, hi = nb - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (src >= bounds[mid]) lo = mid;
    else hi = mid - 1;
  }
  return (uint32_t)lo;
}

// lo/hi: src-range filter for the GROUPED build (the (block,row) slot
// table at 32 MB windows is sb*vp entries — 2^35 at RMAT-29 on one GPU;
// processing groups of windows keeps the transient table inside free
// HBM at the cost of one extra edge pass per group). bounds here is the
// GROUP's boundary slice; block indices are group-local.
__global__ void blocked_count_kernel(uint64_t ep, const V_ID* col,
                                     const E_ID* row_ptr_loc, V_ID vp,
                                     const V_ID* bounds, int nb,
                                     V_ID lo, V_ID hi, uint32_t* counts) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t j = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; j < ep;
       j += stride) {
    V_ID c = col[j];
    if (c < lo || c >= hi) continue;
    V_ID v = row_of_edge(row_ptr_loc, vp, j);
    uint32_t b = block_of(c, bounds, nb);
    atomicAdd(&counts[(uint64_t)b * vp + v], 1u);
  }
}

__global__ void blocked_scatter_kernel(uint64_t ep, const V_ID* col,
                                       const E_ID* row_ptr_loc, V_ID vp,
                                       const V_ID* bounds, int nb,
                                       V_ID lo, V_ID hi,
                                       unsigned long long* cursor,
                                       V_ID* out_col) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t j = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; j < ep;
       j += stride) {
    V_ID c = col[j];
    if (c < lo || c >= hi) continue;
    V_ID v = row_of_edge(row_ptr_loc, vp, j);
    uint32_t b = block_of(c, bounds, nb);
    unsigned long long pos = atomicAdd(&cursor[(uint64_t)b * vp + v], 1ull);
    out_col[pos] = c;
  }
}

// ---------------- local row_ptr from global col_end slice ----------------

__global__ void local_row_ptr_kernel(uint32_t vp, E_ID col_left,
                                     const E_ID* col_end_slice,
                                     E_ID* row_ptr_loc) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i <= vp;
       i += stride) {
    row_ptr_loc[i] = i == 0 ? 0 : col_end_slice[i - 1] - col_left;
  }
}

}  // namespace lux

// ---------------- C ABI ----------------

using namespace lux;

extern "C" {

void lux_gpu_rmat_edges_chunk(uint64_t stream, uint64_t seed, int scale,
                              uint64_t e0, uint64_t ne, V_ID* src,
                              V_ID* dst) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(rmat_edges_kernel, dim3(grid_for(ne)), dim3(BLOCK), 0, s,
                     seed, scale, e0, ne, src, dst);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_rmat_edges(uint64_t stream, uint64_t seed, int scale,
                        uint64_t ne, V_ID* src, V_ID* dst) {
  lux_gpu_rmat_edges_chunk(stream, seed, scale, 0, ne, src, dst);
}

void lux_gpu_rmat_edges_folded_chunk(uint64_t stream, uint64_t seed,
                                     int scale, V_ID nv, uint64_t e0,
                                     uint64_t ne, V_ID* src, V_ID* dst) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(rmat_edges_folded_kernel, dim3(grid_for(ne)),
                     dim3(BLOCK), 0, s, seed, scale, nv, e0, ne, src, dst);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_rmat_edges_folded(uint64_t stream, uint64_t seed, int scale,
                               V_ID nv, uint64_t ne, V_ID* src, V_ID* dst) {
  lux_gpu_rmat_edges_folded_chunk(stream, seed, scale, nv, 0, ne, src, dst);
}

void lux_gpu_bipartite_edges_chunk(uint64_t stream, uint64_t seed,
                                   V_ID n_users, V_ID n_items, uint64_t e0,
                                   uint64_t ne, V_ID* src, V_ID* dst,
                                   WeightType* w) {
  hipStream_t s = (hipStream_t)stream;
  int item_scale = 0;
  while (((V_ID)1 << item_scale) < n_items) item_scale++;
  hipLaunchKernelGGL(bipartite_edges_kernel, dim3(grid_for(ne)), dim3(BLOCK),
                     0, s, seed, n_users, n_items, item_scale, e0, ne, src,
                     dst, w);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_bipartite_edges(uint64_t stream, uint64_t seed, V_ID n_users,
                             V_ID n_items, uint64_t ne, V_ID* src, V_ID* dst,
                             WeightType* w) {
  lux_gpu_bipartite_edges_chunk(stream, seed, n_users, n_items, 0, ne, src,
                                dst, w);
}

void lux_gpu_slice_scatter(uint64_t stream, uint64_t n, const V_ID* src,
                           const V_ID* dst, const WeightType* w, V_ID rl,
                           V_ID rr, unsigned long long* cursor, V_ID* out_col,
                           WeightType* out_w) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(slice_scatter_kernel, dim3(grid_for(n)), dim3(BLOCK), 0,
                     s, n, src, dst, w, rl, rr, cursor, out_col, out_w);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_hist_u32(uint64_t stream, uint64_t n, const V_ID* ids,
                      uint32_t* hist /*pre-zeroed*/) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(hist_u32_kernel, dim3(grid_for(n)), dim3(BLOCK), 0, s, n,
                     ids, hist);
  LUX_POST_LAUNCH(stream);
}

uint32_t lux_gpu_scan_partials_size(uint64_t n) {
  return (uint32_t)((n + (uint64_t)SCAN_TILE - 1) / SCAN_TILE);
}

// n is u64: the blocked-CSC build scans src_blocks*vp counters, which
// exceeds 2^32 at RMAT-28 with 32 MB windows (VERDICT r1 missing #6 — the
// u32 guard forced 128 MB windows there, 69.4 vs a target >=85 GTEPS).
void lux_gpu_scan_end_offsets(uint64_t stream, uint64_t n, const uint32_t* in,
                              E_ID* out_end, unsigned long long* partials) {
  scan_u32_to_end_u64((hipStream_t)stream, n, in, out_end, partials);
  LUX_POST_LAUNCH(stream);
}

// u64 -> u32 narrowing copy (per-block row offsets fit u32: block-local
// edge counts < 2^32) for the halved-row-traffic pull path.
__global__ void u64_to_u32_kernel(uint64_t n, const unsigned long long* in,
                                  uint32_t* out) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint32_t)in[i];
}

void lux_gpu_u64_to_u32(uint64_t stream, uint64_t n,
                        const unsigned long long* in, uint32_t* out) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(u64_to_u32_kernel, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     n, in, out);
  LUX_POST_LAUNCH(stream);
}

// Full CSC build from device edge lists. hist must be pre-zeroed u32[nv];
// cursor u64[nv]; partials u64[lux_gpu_scan_partials_size(nv)].
void lux_gpu_edges_to_csc(uint64_t stream, uint32_t nv, uint64_t ne,
                          const V_ID* src, const V_ID* dst,
                          const WeightType* w, E_ID* col_end, V_ID* out_src,
                          WeightType* out_w, uint32_t* hist,
                          unsigned long long* cursor,
                          unsigned long long* partials) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(hist_u32_kernel, dim3(grid_for(ne)), dim3(BLOCK), 0, s,
                     ne, dst, hist);
  scan_u32_to_end_u64(s, nv, hist, col_end, partials);
  hipLaunchKernelGGL(init_cursor_kernel, dim3(grid_for(nv)), dim3(BLOCK), 0,
                     s, nv, col_end, cursor);
  hipLaunchKernelGGL(scatter_kernel, dim3(grid_for(ne)), dim3(BLOCK), 0, s,
                     ne, src, dst, w, cursor, out_src, out_w);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_blocked_count(uint64_t stream, uint64_t ep, const V_ID* col,
                           const E_ID* row_ptr_loc, V_ID vp,
                           const V_ID* bounds /*device u32[nb+1]*/, int nb,
                           V_ID lo, V_ID hi,
                           uint32_t* counts /*pre-zeroed u32[nb*vp]*/) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(blocked_count_kernel, dim3(grid_for(ep)), dim3(BLOCK),
                     0, s, ep, col, row_ptr_loc, vp, bounds, nb, lo, hi,
                     counts);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_blocked_scatter(uint64_t stream, uint64_t ep, const V_ID* col,
                             const E_ID* row_ptr_loc, V_ID vp,
                             const V_ID* bounds, int nb, V_ID lo, V_ID hi,
                             unsigned long long* cursor, V_ID* out_col) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(blocked_scatter_kernel, dim3(grid_for(ep)), dim3(BLOCK),
                     0, s, ep, col, row_ptr_loc, vp, bounds, nb, lo, hi,
                     cursor, out_col);
  LUX_POST_LAUNCH(stream);
}

void lux_gpu_local_row_ptr(uint64_t stream, uint32_t vp, E_ID col_left,
                           const E_ID* col_end_slice, E_ID* row_ptr_loc) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(local_row_ptr_kernel, dim3(grid_for((uint64_t)vp + 1)),
                     dim3(BLOCK), 0, s, vp, col_left, col_end_slice,
                     row_ptr_loc);
  LUX_POST_LAUNCH(stream);
}

}  // extern "C"
