// Shared device utilities for the gfx950 kernels.
//
// CDNA4 ground rules applied throughout (per the MI355X programming guide):
// 64-wide wavefronts (every shuffle/ballot idiom uses width 64), blocks are
// multiples of 64 (256 = 4 waves default), memory-bound kernels use
// grid-stride loops with the grid capped so the 256-CU chip is oversubscribed
// but not launch-bound, atomics are device-scope (cross-XCD safe by default).
#pragma once
#include <hip/hip_runtime.h>

#include <cstdio>

#include "lux/types.h"

#define LUX_CHECK_HIP(cmd)                                                  \
  do {                                                                      \
    hipError_t e_ = (cmd);                                                  \
    if (e_ != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s:%d: %s\n", __FILE__, __LINE__,          \
              hipGetErrorString(e_));                                       \
      abort();                                                              \
    }                                                                       \
  } while (0)

// Post-launch check for every C-ABI entry point: catches bad launch configs
// immediately; with LUX_SYNC_CHECK=1 in the env also synchronises and
// surfaces async faults at the offending call (debug mode).
// NOTE: hipGetLastError() is process-global sticky state — other libraries
// (torch probing) can leave a stale error behind, so the non-sync path only
// warns (once). LUX_SYNC_CHECK=1 turns every entry point into a synchronous
// checkpoint that aborts at the offending call (debug mode).
#define LUX_POST_LAUNCH(stream)                                             \
  do {                                                                      \
    hipError_t e_ = hipGetLastError();                                      \
    if (e_ != hipSuccess) {                                                 \
      static int warned_ = 0;                                               \
      if (!warned_++)                                                       \
        fprintf(stderr,                                                     \
                "[lux] warning: HIP error state at %s: %s (may be "         \
                "pre-existing; set LUX_SYNC_CHECK=1 to localise)\n",        \
                __func__, hipGetErrorString(e_));                           \
    }                                                                       \
    if (getenv("LUX_SYNC_CHECK")) {                                         \
      e_ = hipStreamSynchronize((hipStream_t)(stream));                     \
      if (e_ != hipSuccess) {                                               \
        fprintf(stderr, "HIP async error %s:%d (%s): %s\n", __FILE__,       \
                __LINE__, __func__, hipGetErrorString(e_));                 \
        abort();                                                            \
      }                                                                     \
    }                                                                       \
  } while (0)

namespace lux {

constexpr int WAVE = 64;
constexpr int BLOCK = 256;            // 4 waves
constexpr int MAX_GRID = 8192;        // grid-stride beyond this

__host__ __device__ __forceinline__ uint32_t ceil_div_u32(uint32_t a,
                                                          uint32_t b) {
  return (a + b - 1) / b;
}
__host__ __forceinline__ int grid_for(uint64_t work, int block = BLOCK) {
  uint64_t g = (work + block - 1) / block;
  return (int)(g > MAX_GRID ? MAX_GRID : (g == 0 ? 1 : g));
}

// Wave-wide (64-lane) reductions.
template <typename T>
__device__ __forceinline__ T wave_reduce_sum(T v) {
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;
}
__device__ __forceinline__ uint32_t wave_reduce_min(uint32_t v) {
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    uint32_t o = __shfl_down(v, off, WAVE);
    v = o < v ? o : v;
  }
  return v;
}
__device__ __forceinline__ uint32_t wave_reduce_max(uint32_t v) {
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    uint32_t o = __shfl_down(v, off, WAVE);
    v = o > v ? o : v;
  }
  return v;
}

// Block-wide (BLOCK=256, 4 waves) sum via one LDS slot per wave.
template <typename T>
__device__ __forceinline__ T block_reduce_sum(T v, T* lds4) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds4[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = (lane < (int)(blockDim.x / WAVE)) ? lds4[lane] : T(0);
    v = wave_reduce_sum(v);
  }
  return v;  // valid in wave 0
}

// Block-wide exclusive scan over one value per thread (u32), BLOCK<=1024.
// Returns this thread's exclusive prefix; *total gets the block sum.
// Classic LDS ladder — replaces the reference's cub::BlockScan usage
// (e.g. sssp_gpu.cu:94,148) with a wave64-shaped hand-rolled scan.
template <typename T, int NT>
__device__ __forceinline__ T block_exscan(T v, T* lds, T* total) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x >> 6;
  // in-wave inclusive scan via shfl_up
  T x = v;
  for (int off = 1; off < WAVE; off <<= 1) {
    T y = __shfl_up(x, off, WAVE);
    if (lane >= off) x += y;
  }
  constexpr int NW = NT / WAVE;
  if (lane == WAVE - 1) lds[wid] = x;
  __syncthreads();
  if (threadIdx.x == 0) {
    T run = 0;
    for (int w = 0; w < NW; w++) {
      T t = lds[w];
      lds[w] = run;
      run += t;
    }
    lds[NW] = run;
  }
  __syncthreads();
  T out = x - v + lds[wid];
  if (total) *total = lds[NW];
  return out;
}

}  // namespace lux
