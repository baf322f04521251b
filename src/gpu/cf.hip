// Collaborative filtering (matrix factorization SGD sweep) on gfx950.
//
// Semantics per reference cf_kernel (colfilter_gpu.cu:32-104): per dst
// vertex, over its in-edges with OLD vectors: err = w - <src_vec, dst_vec>;
// acc += err * src_vec; new = old + GAMMA * (acc - LAMBDA * old). K (latent
// rank) is a runtime parameter (reference: 20; benchmark config: 64),
// K <= 256.
//
// MI355X mapping: one 64-lane wave per dst vertex, lane l owning latent dims
// l, l+64, ... — src-vector gathers are naturally coalesced (64 lanes x 4 B
// = one 256 B contiguous vector read per edge); the edge dot product is a
// wave shuffle-reduce. Hub vertices (deg >= T2) are split into edge chunks
// (one 4-wave block per chunk) accumulating partials into the output buffer
// with float atomics, then a vector epilogue applies the update — reusing
// the degree bins built by pull.hip (same thresholds, same lists).
#include "gpu_common.h"

namespace lux {

constexpr int CF_MAXC = 4;  // K <= 4*64

// Wave-private LDS ordering: make this wave's ds_writes visible to its own
// cross-lane ds_reads without a block barrier (waves in a block work on
// independent vertices, so __syncthreads() would mismatch). Lockstep wave +
// lgkmcnt(0) is sufficient; "memory" stops compiler reordering.
__device__ __forceinline__ void wave_lds_sync() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

struct CFArgs {
  const E_ID* row_ptr;     // u64[vp+1] local
  const V_ID* col;         // u32[ep]
  const WeightType* w;     // i32[ep]
  const float* oldv;       // f32[nv*K]
  float* newv;             // f32[vp*K]
  V_ID row_left;
  int K;
};

__device__ __forceinline__ float wave_bcast(float v, int src_lane) {
  return __shfl(v, src_lane, WAVE);
}

// one wave per dst vertex (bin0+bin1 lists)
__global__ void cf_wave_kernel(uint32_t n, const V_ID* binlist, CFArgs a) {
  int lane = threadIdx.x & (WAVE - 1);
  uint64_t wave_id = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  uint64_t nwaves = ((uint64_t)gridDim.x * blockDim.x) / WAVE;
  int nc = (a.K + WAVE - 1) / WAVE;
  for (uint64_t i = wave_id; i < n; i += nwaves) {
    V_ID v = binlist[i];
    E_ID b = a.row_ptr[v], e = a.row_ptr[v + 1];
    const float* dv_p = a.oldv + (uint64_t)(a.row_left + v) * a.K;
    float dv[CF_MAXC], acc[CF_MAXC];
#pragma unroll
    for (int c = 0; c < CF_MAXC; c++) {
      int k = c * WAVE + lane;
      dv[c] = (c < nc && k < a.K) ? dv_p[k] : 0.0f;
      acc[c] = 0.0f;
    }
    for (E_ID j = b; j < e; j++) {
      const float* sv_p = a.oldv + (uint64_t)a.col[j] * a.K;
      float sv[CF_MAXC];
      float dot = 0.0f;
#pragma unroll
      for (int c = 0; c < CF_MAXC; c++) {
        int k = c * WAVE + lane;
        sv[c] = (c < nc && k < a.K) ? sv_p[k] : 0.0f;
        dot += sv[c] * dv[c];
      }
      dot = wave_reduce_sum(dot);
      float err = (float)a.w[j] - wave_bcast(dot, 0);
#pragma unroll
      for (int c = 0; c < CF_MAXC; c++) acc[c] += err * sv[c];
    }
    // output pre-seeded by the engine with old*(1 - GAMMA*LAMBDA); sweeps
    // add GAMMA*acc (identical update to colfilter_gpu.cu:96-102, split so
    // deg-0 rows and hub chunks need no separate epilogue)
    float* out = a.newv + (uint64_t)v * a.K;
#pragma unroll
    for (int c = 0; c < CF_MAXC; c++) {
      int k = c * WAVE + lane;
      if (c < nc && k < a.K) out[k] += CF_GAMMA * acc[c];
    }
  }
}

// ---------------------------------------------------------------------------
// K<=64 fast path: LDS-tiled wave kernel.
//
// The generic wave kernel above serialises a ~500-cycle dependent chain per
// edge (gather -> dot -> 6-step shuffle reduce -> broadcast -> axpy). Here a
// wave stages TILE (32|64) edges' src vectors in an LDS tile S[TILE][65], a
// coalesced lane=dim load, then switches lane mapping twice:
//   dot phase    lane=edge: each lane walks its OWN row serially against the
//                dst vector (LDS broadcast reads) -> 64 independent FMA
//                chains instead of 64 serialised wave reduces;
//   update phase lane=dim:  acc_k += err_e * S[e][k], err_e broadcast.
// Row stride 65 words makes both access patterns bank-conflict-free
// ((e+k) mod 32 distinct within each 32-lane group).
// ---------------------------------------------------------------------------
constexpr int CF_ROW = 65;  // dim pitch: 64 dims + bank pad
constexpr int CF_TB = 64;   // one wave per workgroup

// TILE = edges staged per LDS tile. 64: 17.2 KB/wave (9 wg/CU); 32:
// 8.8 KB/wave (16 wg/CU cap) — more waves to queue LLC gathers, but the
// lane=edge dot phase runs half-idle. A/B via LUX_CF_TILE=32|64.
template <int TILE> struct CFTileLds {
  float S[TILE * CF_ROW];
  float dv[64];
  float err[TILE];
};

// Stage up to CF_TILE edges [t, t+rem) into lds->S and per-lane col/weight;
// returns this lane's weight (lane e holds edge t+e's weight).
//
// The load loop MUST be statically unrolled in depth-16 batches: with a
// dynamic trip count the compiler keeps it rolled and each ds_write waits
// for its own global load (one outstanding ~500-cycle LLC access per edge —
// measured 35.9 ms/sweep on the NetFlix config, WORSE than the un-tiled v1).
// Batching 16 loads into registers before the LDS writes keeps 16 reads in
// flight per wave.
template <int TILE>
__device__ __forceinline__ float cf_stage_tile(CFTileLds<TILE>* lds, E_ID t,
                                               int rem, const CFArgs& a,
                                               int lane) {
  uint32_t mycol = 0;
  float myw = 0.0f;
  if (lane < rem) {
    mycol = a.col[t + lane];
    myw = (float)a.w[t + lane];
  }
  if (rem == TILE) {
#pragma unroll
    for (int r0 = 0; r0 < TILE; r0 += 16) {
      float tmp[16];
#pragma unroll
      for (int r = 0; r < 16; r++) {
        uint32_t src = __shfl(mycol, r0 + r, WAVE);
        tmp[r] = lane < a.K ? a.oldv[(uint64_t)src * a.K + lane] : 0.0f;
      }
#pragma unroll
      for (int r = 0; r < 16; r++)
        lds->S[(r0 + r) * CF_ROW + lane] = tmp[r];
    }
    return myw;
  }
  int r = 0;
  for (; r + 4 <= rem; r += 4) {
    float tmp[4];
#pragma unroll
    for (int q = 0; q < 4; q++) {
      uint32_t src = __shfl(mycol, r + q, WAVE);
      tmp[q] = lane < a.K ? a.oldv[(uint64_t)src * a.K + lane] : 0.0f;
    }
#pragma unroll
    for (int q = 0; q < 4; q++) lds->S[(r + q) * CF_ROW + lane] = tmp[q];
  }
  for (; r < rem; r++) {
    uint32_t src = __shfl(mycol, r, WAVE);
    lds->S[r * CF_ROW + lane] =
        lane < a.K ? a.oldv[(uint64_t)src * a.K + lane] : 0.0f;
  }
  return myw;
}

// Full-tile (rem==TILE), K==64 pass: every loop statically unrolled.
template <int TILE>
__device__ __forceinline__ float cf_tile_pass_fast(CFTileLds<TILE>* lds,
                                                   float myw, float acc,
                                                   int lane) {
  wave_lds_sync();
  if (lane < TILE) {  // lane = edge; rows only exist for lane < TILE
    const float* row = &lds->S[lane * CF_ROW];
    float d0 = 0, d1 = 0, d2 = 0, d3 = 0;
#pragma unroll
    for (int k = 0; k < 64; k += 4) {
      d0 += row[k] * lds->dv[k];
      d1 += row[k + 1] * lds->dv[k + 1];
      d2 += row[k + 2] * lds->dv[k + 2];
      d3 += row[k + 3] * lds->dv[k + 3];
    }
    lds->err[lane] = myw - ((d0 + d1) + (d2 + d3));
  }
  wave_lds_sync();
  float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
#pragma unroll
  for (int r = 0; r < TILE; r += 4) {
    a0 += lds->err[r] * lds->S[r * CF_ROW + lane];
    a1 += lds->err[r + 1] * lds->S[(r + 1) * CF_ROW + lane];
    a2 += lds->err[r + 2] * lds->S[(r + 2) * CF_ROW + lane];
    a3 += lds->err[r + 3] * lds->S[(r + 3) * CF_ROW + lane];
  }
  wave_lds_sync();
  return acc + ((a0 + a1) + (a2 + a3));
}

// dot+err+update over a staged tile; returns updated acc (lane=dim).
template <int TILE>
__device__ __forceinline__ float cf_tile_pass(CFTileLds<TILE>* lds, int rem,
                                              float myw, int K, float acc,
                                              int lane) {
  wave_lds_sync();  // S rows visible
  if (lane < TILE) {  // lane = edge; rows only exist for lane < TILE
    float d0 = 0, d1 = 0, d2 = 0, d3 = 0;
    const float* row = &lds->S[lane * CF_ROW];
    int k = 0;
    for (; k + 3 < K; k += 4) {
      d0 += row[k] * lds->dv[k];
      d1 += row[k + 1] * lds->dv[k + 1];
      d2 += row[k + 2] * lds->dv[k + 2];
      d3 += row[k + 3] * lds->dv[k + 3];
    }
    for (; k < K; k++) d0 += row[k] * lds->dv[k];
    float dot = (d0 + d1) + (d2 + d3);
    lds->err[lane] = lane < rem ? myw - dot : 0.0f;
  }
  wave_lds_sync();  // err visible
  float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
  int r = 0;
  for (; r + 3 < rem; r += 4) {
    a0 += lds->err[r] * lds->S[r * CF_ROW + lane];
    a1 += lds->err[r + 1] * lds->S[(r + 1) * CF_ROW + lane];
    a2 += lds->err[r + 2] * lds->S[(r + 2) * CF_ROW + lane];
    a3 += lds->err[r + 3] * lds->S[(r + 3) * CF_ROW + lane];
  }
  for (; r < rem; r++) a0 += lds->err[r] * lds->S[r * CF_ROW + lane];
  wave_lds_sync();  // tile consumed before the next stage overwrites it
  return acc + ((a0 + a1) + (a2 + a3));
}

// accumulate gradient over edge range [b, e); lane = dim
template <int TILE>
__device__ __forceinline__ float cf_range_acc(CFTileLds<TILE>* lds, E_ID b,
                                              E_ID e, const CFArgs& a,
                                              int lane) {
  float acc = 0.0f;
  E_ID t = b;
  if (a.K == 64) {
    for (; t + TILE <= e; t += TILE) {
      float myw = cf_stage_tile<TILE>(lds, t, TILE, a, lane);
      acc = cf_tile_pass_fast<TILE>(lds, myw, acc, lane);
    }
  }
  for (; t < e; t += TILE) {
    int rem = (int)(e - t < TILE ? e - t : (E_ID)TILE);
    float myw = cf_stage_tile<TILE>(lds, t, rem, a, lane);
    acc = cf_tile_pass<TILE>(lds, rem, myw, a.K, acc, lane);
  }
  return acc;
}

// one wave per dst vertex over a bin list (deg < T2)
template <int TILE>
__global__ __launch_bounds__(CF_TB) void cf_tile_kernel(uint32_t n,
                                                        const V_ID* binlist,
                                                        CFArgs a) {
  __shared__ CFTileLds<TILE> lds;
  int lane = threadIdx.x;
  uint64_t nwaves = gridDim.x;
  for (uint64_t i = blockIdx.x; i < n; i += nwaves) {
    V_ID v = binlist[i];
    E_ID b = a.row_ptr[v], e = a.row_ptr[v + 1];
    lds.dv[lane] =
        lane < a.K ? a.oldv[(uint64_t)(a.row_left + v) * a.K + lane] : 0.0f;
    float acc = cf_range_acc<TILE>(&lds, b, e, a, lane);
    if (lane < a.K) a.newv[(uint64_t)v * a.K + lane] += CF_GAMMA * acc;
  }
}

// one wave per hub chunk (deg >= T2), atomic epilogue
template <int TILE>
__global__ __launch_bounds__(CF_TB) void cf_tile_chunk_kernel(
    uint32_t n2, const uint2* bin2, V_ID chunk_edges, CFArgs a) {
  __shared__ CFTileLds<TILE> lds;
  int lane = threadIdx.x;
  for (uint32_t i = blockIdx.x; i < n2; i += gridDim.x) {
    uint2 ent = bin2[i];
    V_ID v = ent.x;
    E_ID b = a.row_ptr[v] + (E_ID)ent.y * chunk_edges;
    E_ID e = a.row_ptr[v + 1];
    if (e > b + chunk_edges) e = b + chunk_edges;
    lds.dv[lane] =
        lane < a.K ? a.oldv[(uint64_t)(a.row_left + v) * a.K + lane] : 0.0f;
    float acc = cf_range_acc<TILE>(&lds, b, e, a, lane);
    if (lane < a.K)
      atomicAdd(&a.newv[(uint64_t)v * a.K + lane], CF_GAMMA * acc);
  }
}

__global__ void cf_chunk_kernel(uint32_t n2, const uint2* bin2,
                                V_ID chunk_edges, CFArgs a) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x >> 6;
  int waves = blockDim.x / WAVE;
  int nc = (a.K + WAVE - 1) / WAVE;
  for (uint32_t i = blockIdx.x; i < n2; i += gridDim.x) {
    uint2 ent = bin2[i];
    V_ID v = ent.x;
    E_ID b = a.row_ptr[v] + (E_ID)ent.y * chunk_edges;
    E_ID e = a.row_ptr[v + 1];
    if (e > b + chunk_edges) e = b + chunk_edges;
    const float* dv_p = a.oldv + (uint64_t)(a.row_left + v) * a.K;
    float dv[CF_MAXC], acc[CF_MAXC];
#pragma unroll
    for (int c = 0; c < CF_MAXC; c++) {
      int k = c * WAVE + lane;
      dv[c] = (c < nc && k < a.K) ? dv_p[k] : 0.0f;
      acc[c] = 0.0f;
    }
    // waves interleave over the chunk's edges
    for (E_ID j = b + wid; j < e; j += waves) {
      const float* sv_p = a.oldv + (uint64_t)a.col[j] * a.K;
      float sv[CF_MAXC];
      float dot = 0.0f;
#pragma unroll
      for (int c = 0; c < CF_MAXC; c++) {
        int k = c * WAVE + lane;
        sv[c] = (c < nc && k < a.K) ? sv_p[k] : 0.0f;
        dot += sv[c] * dv[c];
      }
      dot = wave_reduce_sum(dot);
      float err = (float)a.w[j] - wave_bcast(dot, 0);
#pragma unroll
      for (int c = 0; c < CF_MAXC; c++) acc[c] += err * sv[c];
    }
    float* out = a.newv + (uint64_t)v * a.K;
#pragma unroll
    for (int c = 0; c < CF_MAXC; c++) {
      int k = c * WAVE + lane;
      if (c < nc && k < a.K) atomicAdd(&out[k], CF_GAMMA * acc[c]);
    }
  }
}


// SGD sweep output seed: new = old * (1 - GAMMA*LAMBDA); the sweep kernels
// then ADD GAMMA*acc (colfilter_gpu.cu:96-102 split, see cf_wave_kernel).
__global__ void cf_seed_kernel(uint64_t n, const float* oldv, float* newv) {
  uint64_t stride = (uint64_t)blockDim.x * gridDim.x;
  const float f = 1.0f - CF_GAMMA * CF_LAMBDA;
  for (uint64_t i = blockIdx.x * (uint64_t)blockDim.x + threadIdx.x; i < n;
       i += stride)
    newv[i] = oldv[i] * f;
}

}  // namespace lux

using namespace lux;

extern "C" {

void lux_gpu_cf_seed(uint64_t stream, uint64_t n, const float* oldv,
                     float* newv) {
  hipStream_t s = (hipStream_t)stream;
  hipLaunchKernelGGL(cf_seed_kernel, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     n, oldv, newv);
  LUX_POST_LAUNCH(stream);
}

// One CF sweep over my partition. Uses the pull.hip bin lists (bin0 and
// bin1 both go to the wave kernel; bin2 chunked, chunk_edges = 8192).
void lux_gpu_cf_iter(uint64_t stream, uint32_t n0, const V_ID* bin0,
                     uint32_t n1, const V_ID* bin1, uint32_t n2,
                     const uint2* bin2, uint32_t nbig, const V_ID* bin2v,
                     const E_ID* row_ptr, const V_ID* col,
                     const WeightType* w, const float* oldv, float* newv,
                     V_ID row_left, int K) {
  hipStream_t s = (hipStream_t)stream;
  CFArgs a{row_ptr, col, w, oldv, newv, row_left, K};
  if (K <= 64) {
    // LDS-tiled fast path (benchmark config K=64; reference K=20)
    static int tile = [] {  // 32 measured best (8.6 vs 9.8 ms/sweep
      const char* t = getenv("LUX_CF_TILE");  // NetFlix K=64: 16 wg/CU
      return t && atoi(t) == 64 ? 64 : 32;    // beats 9 despite idle lanes)
    }();
    if (nbig) {
      dim3 g(n2 > MAX_GRID ? MAX_GRID : n2);
      if (tile == 32)
        hipLaunchKernelGGL(cf_tile_chunk_kernel<32>, g, dim3(CF_TB), 0, s,
                           n2, bin2, (V_ID)8192, a);
      else
        hipLaunchKernelGGL(cf_tile_chunk_kernel<64>, g, dim3(CF_TB), 0, s,
                           n2, bin2, (V_ID)8192, a);
    }
    if (n1) {
      dim3 g(n1 > MAX_GRID ? MAX_GRID : n1);
      if (tile == 32)
        hipLaunchKernelGGL(cf_tile_kernel<32>, g, dim3(CF_TB), 0, s, n1,
                           bin1, a);
      else
        hipLaunchKernelGGL(cf_tile_kernel<64>, g, dim3(CF_TB), 0, s, n1,
                           bin1, a);
    }
    if (n0)
      hipLaunchKernelGGL(cf_wave_kernel, dim3(grid_for((uint64_t)n0 * WAVE)),
                         dim3(BLOCK), 0, s, n0, bin0, a);
    LUX_POST_LAUNCH(stream);
    return;
  }
  if (nbig) {
    hipLaunchKernelGGL(cf_chunk_kernel, dim3(n2 > MAX_GRID ? MAX_GRID : n2),
                       dim3(BLOCK), 0, s, n2, bin2, (V_ID)8192, a);
  }
  if (n1)
    hipLaunchKernelGGL(cf_wave_kernel, dim3(grid_for((uint64_t)n1 * WAVE)),
                       dim3(BLOCK), 0, s, n1, bin1, a);
  if (n0)
    hipLaunchKernelGGL(cf_wave_kernel, dim3(grid_for((uint64_t)n0 * WAVE)),
                       dim3(BLOCK), 0, s, n0, bin0, a);
  LUX_POST_LAUNCH(stream);
}

}  // extern "C"
