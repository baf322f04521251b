// Deterministic, embarrassingly-parallel R-MAT edge generator.
//
// Shared between the CPU core (g++) and the HIP kernels (hipcc device code):
// edge e of a (seed, scale, ne) instance is a pure function of (seed, e), so
// the CPU and GPU generators produce bit-identical edge lists — the GPU
// graph builder is validated against the CPU one in tests/test_rmat.py.
//
// The reference ships no generator (graphs arrive as .lux files via
// tools/converter.cc); BASELINE.md's synthetic RMAT-16/27 and shaped graphs
// are produced by this module.
#pragma once
#include "types.h"

#if defined(__HIPCC__)
#define LUX_HD __host__ __device__ __forceinline__
#else
#define LUX_HD inline
#endif

namespace lux {

LUX_HD uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// R-MAT quadrant probabilities (Graph500 defaults).
constexpr float RMAT_A = 0.57f;
constexpr float RMAT_B = 0.19f;
constexpr float RMAT_C = 0.19f;

// Generate directed edge e (src, dst both in [0, 2^scale)). Self-loops and
// duplicate edges are kept, matching how RMAT-27 in the reference's dataset
// table reaches exactly 2^31 edges (README.md:79-86).
LUX_HD void rmat_edge(uint64_t seed, uint64_t e, int scale, V_ID* src,
                      V_ID* dst) {
  uint64_t s = splitmix64(seed ^ (e * 0xD1B54A32D192ED03ull));
  V_ID u = 0, v = 0;
  for (int bit = 0; bit < scale; bit++) {
    // Two fresh 16-bit draws per level from an evolving splitmix stream.
    s = splitmix64(s);
    uint32_t r = (uint32_t)(s & 0xFFFFFFFFu);
    float p = (float)(r & 0xFFFF) * (1.0f / 65536.0f);
    uint32_t hi_u, hi_v;
    if (p < RMAT_A) {
      hi_u = 0; hi_v = 0;
    } else if (p < RMAT_A + RMAT_B) {
      hi_u = 0; hi_v = 1;
    } else if (p < RMAT_A + RMAT_B + RMAT_C) {
      hi_u = 1; hi_v = 0;
    } else {
      hi_u = 1; hi_v = 1;
    }
    u = (u << 1) | hi_u;
    v = (v << 1) | hi_v;
  }
  *src = u;
  *dst = v;
}

// Uniform random edge (for Twitter-/NetFlix-shaped synthetic graphs a
// skew-free generator is wrong; bench uses rmat_edge with the scale rounded
// up and ids folded by modulo — see uniform_fold below).
LUX_HD V_ID fold_id(V_ID x, V_ID nv) { return (nv & (nv - 1)) == 0 ? (x & (nv - 1)) : (x % nv); }

// RMAT with a non-power-of-two vertex count (Twitter-/NetFlix-shaped
// synthetic graphs): generate at the ceiling scale, fold ids by modulo.
// Skew is preserved (fold maps the hot low-id region onto itself).
LUX_HD void rmat_edge_folded(uint64_t seed, uint64_t e, int scale, V_ID nv,
                             V_ID* src, V_ID* dst) {
  rmat_edge(seed, e, scale, src, dst);
  *src = fold_id(*src, nv);
  *dst = fold_id(*dst, nv);
}

// Deterministic edge weight for weighted synthetic graphs (CF): int in
// [1, 5] like a ratings matrix.
LUX_HD WeightType rmat_weight(uint64_t seed, uint64_t e) {
  return (WeightType)(splitmix64(seed ^ 0xABCD1234u ^ (e * 0x2545F4914F6CDD1Dull)) % 5) + 1;
}

// Bipartite synthetic generator for CF (NetFlix-shaped: users x items).
// Edge pair 2r/2r+1 stores rating r in BOTH directions (user <-> item),
// matching the reference's NetFlix file whose 200.9M edges are 100.5M
// ratings x 2 — the pull-model CF sweeps then update BOTH sides (items
// gather user vectors and vice versa; a one-directional store would
// freeze the user factors forever). Items are RMAT-popularity-skewed,
// users hash-uniform; both directions of a rating share the weight
// (rmat_weight(seed, e >> 1)).
LUX_HD void bipartite_edge(uint64_t seed, uint64_t e, V_ID n_users,
                           V_ID n_items, int item_scale, V_ID* src,
                           V_ID* dst) {
  uint64_t r = e >> 1;
  V_ID iu, iv;
  rmat_edge(seed, r, item_scale, &iu, &iv);
  uint64_t s = splitmix64(seed ^ 0x5EEDF00Dull ^ (r * 0x9E3779B97F4A7C15ull));
  V_ID user = fold_id((V_ID)s, n_users);
  V_ID item = n_users + fold_id(iv, n_items);
  if (e & 1) {
    *src = item;
    *dst = user;
  } else {
    *src = user;
    *dst = item;
  }
}

}  // namespace lux
