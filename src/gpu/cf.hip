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


}  // namespace lux

using namespace lux;

extern "C" {

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
