// Collaborative filtering — ALS sweep with MFMA Gram accumulation (gfx950).
//
// A second CF optimizer beside the reference-parity SGD sweep (cf.hip;
// reference colfilter_gpu.cu:32-104). One ALS sweep solves, per dst vertex,
// the exact normal equations the SGD sweep only steps toward:
//     (S^T S + lambda I) d = S^T w
// where S (deg x K) stacks the OLD src vectors of v's in-edges and w the
// edge weights. This is the SGD fixed point (grad = S^T(w - S d) - lambda d
// = 0), so both optimizers share semantics; ALS just converges in far fewer
// sweeps. Data access per sweep is identical to the SGD sweep (gather src
// vectors over in-edges, publish my slice), so the distributed exchange
// (all-gather of vector slices) is unchanged.
//
// MI355X mapping: the Gram matrix G = S^T S is the dense hot loop —
// K^2 * deg MACs per vertex — and is exactly MFMA-shaped. One 64-lane wave
// owns one dst vertex, stages ALS_TILE-edge tiles of src vectors in LDS,
// and accumulates the upper-triangular 16x16 tiles of G (K <= 64 -> 4x4
// tile grid, 10 upper tiles) with v_mfma_f32_16x16x4_f32: per 4-edge group,
// fragment f[t] = S[e0 + (lane>>4)][t*16 + (lane&15)] serves as operand A of
// tile-row t AND operand B of tile-col t (A[i][k]=S[e0+k][16ti+i],
// B[k][j]=S[e0+k][16tj+j] per the gfx950 f32 MFMA lane maps), so 4 LDS reads
// feed 10 MFMAs. fp32-input MFMA runs at the fp32 vector rate on gfx950 but
// packs the 64x64 outer-product accumulation into 40 accumulator VGPRs with
// no cross-lane reduction — the vector formulation needs either 64 serial
// wave reduces per 4 edges or an LDS round-trip per rank-1 update.
// The 64x64 SPD solve (Cholesky + two triangular solves) runs in the same
// wave on the LDS copy of G — no global scratch for the common case.
// Hub vertices (deg >= T2, bin2 chunk lists from pull.hip) accumulate
// partial G/rhs into global scratch with atomics, then a second kernel
// solves per hub.
#include "gpu_common.h"

namespace lux {

constexpr int ALS_K = 64;        // max latent rank of the MFMA path
constexpr int ALS_ROW = 65;      // LDS row pitch (bank-conflict pad)
constexpr int ALS_TILE = 32;     // edges staged per LDS tile (32: half the
                                 // LDS per wave -> ~2x waves/CU; same
                                 // occupancy lesson as cf.hip TILE=32)
constexpr int ALS_TB = 64;       // one wave per workgroup

using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ void als_lds_sync() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

struct CFAlsArgs {
  const E_ID* row_ptr;   // u64[vp+1] local
  const V_ID* col;       // u32[ep]
  const WeightType* w;   // i32[ep]
  const float* oldv;     // f32[nv*K]
  const __bf16* oldv_bf; // bf16 gather replica (nullable): halves the
                         // per-edge gather bytes AND the LLC footprint of
                         // the hot src-vector table on the bf16 Gram path
  float* newv;           // f32[vp*K] (pre-seeded with old slice)
  V_ID row_left;
  int K;                 // <= 64
};

struct AlsGramLds {
  float S[ALS_TILE * ALS_ROW];  // staged src vectors, row = edge
  float W[ALS_TILE];            // staged edge weights
};

// bf16 Gram path (default): the fp32 MFMA formulation above runs at the
// 155 TF f32 matrix rate and IS the sweep's bound (measured r2: the gram
// chunk kernel ~14 ms/sweep ~= the f32-MFMA issue floor). bf16 inputs
// with fp32 accumulation move the Gram onto v_mfma_f32_16x16x32_bf16
// (2075 TF): one K=32 MFMA per 16x16 tile per 32-edge LDS tile — 10
// MFMAs per tile instead of 80 — and the rhs folds from registers, so
// the LDS image shrinks to a transposed bf16 [dim][edge] panel whose
// fragments are single ds_read_b128s. LUX_ALS_F32=1 restores the exact
// fp32 path.
constexpr int ALS_BPITCH = 40;  // bf16 row pitch: 80 B = b128-aligned;
                                // 16-lane b128 groups land on distinct
                                // 4-bank runs (20 dwords/row, 20*16%64
                                // spread) -> conflict-free reads

struct alignas(16) AlsGramLdsBf {
  __bf16 S[ALS_K * ALS_BPITCH];  // TRANSPOSED: row = dim, col = edge
  float W[ALS_TILE];
};

struct AlsLds {
  union {
    AlsGramLds g;               // staging (S/W), fp32 path
    AlsGramLdsBf gb;            // staging, bf16 path
  };
  float G[ALS_K * ALS_ROW];     // Gram -> Cholesky factor (in place)
};

// ---- Gram accumulation over one vertex's edge range [b, e) ----
// acc[10] are the upper tiles (ti<=tj) in order (0,0)(0,1)(0,2)(0,3)
// (1,1)(1,2)(1,3)(2,2)(2,3)(3,3); rhs is lane=dim.
//
// Tile-level software pipeline: the NEXT tile's src-vector loads issue into
// registers (tmp[ALS_TILE]) right before the CURRENT tile's MFMA burst
// (~80 x 32-cycle issue at ALS_TILE=32), so their ~LLC latency hides under
// the matrix work; one LDS buffer suffices (regs -> LDS at tile start).

// Issue the whole tile's vector loads into registers (all outstanding).
__device__ __forceinline__ void als_vec_loads(const CFAlsArgs& a,
                                              uint32_t mycol, int rem,
                                              int lane,
                                              float tmp[ALS_TILE]) {
#pragma unroll
  for (int r = 0; r < ALS_TILE; r++) {
    uint32_t src = __shfl(mycol, r, WAVE);
    tmp[r] = (r < rem && lane < a.K)
                 ? a.oldv[(uint64_t)src * a.K + lane]
                 : 0.0f;
  }
}

__device__ __forceinline__ void als_gram_range(const CFAlsArgs& a, E_ID b,
                                               E_ID e, int lane,
                                               AlsGramLds* lds,
                                               f32x4 acc[10], float* rhs) {
  if (b >= e) return;
  int rem = (int)(e - b < ALS_TILE ? e - b : (E_ID)ALS_TILE);
  uint32_t mycol = 0;
  float myw = 0.0f;
  if (lane < rem) {
    mycol = a.col[b + lane];
    myw = (float)a.w[b + lane];
  }
  float tmp[ALS_TILE];
  als_vec_loads(a, mycol, rem, lane, tmp);
  for (E_ID t = b; t < e; t += ALS_TILE) {
    // stage the (prefetched) current tile
    if (lane < ALS_TILE) lds->W[lane] = myw;
#pragma unroll
    for (int r = 0; r < ALS_TILE; r++)
      lds->S[r * ALS_ROW + lane] = tmp[r];
    int rem_cur = rem;
    // prefetch next tile's col/weight now, vectors after the rhs pass
    E_ID t2 = t + ALS_TILE;
    bool more = t2 < e;
    uint32_t ncol = 0;
    float nw = 0.0f;
    if (more) {
      rem = (int)(e - t2 < ALS_TILE ? e - t2 : (E_ID)ALS_TILE);
      if (lane < rem) {
        ncol = a.col[t2 + lane];
        nw = (float)a.w[t2 + lane];
      }
    }
    int rem4 = (rem_cur + 3) & ~3;
    als_lds_sync();
    // rhs += sum_r w_r * S[r][lane] (4 independent LDS-read chains)
    float r0 = 0, r1 = 0, r2 = 0, r3 = 0;
    int rr = 0;
    for (; rr + 4 <= rem_cur; rr += 4) {
      r0 += lds->W[rr] * lds->S[rr * ALS_ROW + lane];
      r1 += lds->W[rr + 1] * lds->S[(rr + 1) * ALS_ROW + lane];
      r2 += lds->W[rr + 2] * lds->S[(rr + 2) * ALS_ROW + lane];
      r3 += lds->W[rr + 3] * lds->S[(rr + 3) * ALS_ROW + lane];
    }
    for (; rr < rem_cur; rr++)
      r0 += lds->W[rr] * lds->S[rr * ALS_ROW + lane];
    *rhs += ((r0 + r1) + (r2 + r3));
    // issue the next tile's vector loads; they land under the MFMA burst
    if (more) {
      myw = nw;
      als_vec_loads(a, ncol, rem, lane, tmp);
      mycol = ncol;
    }
    // MFMA over 4-edge groups
    int erow = lane >> 4, ecol = lane & 15;
    for (int kk = 0; kk < rem4; kk += 4) {
      const float* base = &lds->S[(kk + erow) * ALS_ROW + ecol];
      float f0 = base[0], f1 = base[16], f2 = base[32], f3 = base[48];
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(f0, f0, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x4f32(f0, f1, acc[1], 0, 0, 0);
      acc[2] = __builtin_amdgcn_mfma_f32_16x16x4f32(f0, f2, acc[2], 0, 0, 0);
      acc[3] = __builtin_amdgcn_mfma_f32_16x16x4f32(f0, f3, acc[3], 0, 0, 0);
      acc[4] = __builtin_amdgcn_mfma_f32_16x16x4f32(f1, f1, acc[4], 0, 0, 0);
      acc[5] = __builtin_amdgcn_mfma_f32_16x16x4f32(f1, f2, acc[5], 0, 0, 0);
      acc[6] = __builtin_amdgcn_mfma_f32_16x16x4f32(f1, f3, acc[6], 0, 0, 0);
      acc[7] = __builtin_amdgcn_mfma_f32_16x16x4f32(f2, f2, acc[7], 0, 0, 0);
      acc[8] = __builtin_amdgcn_mfma_f32_16x16x4f32(f2, f3, acc[8], 0, 0, 0);
      acc[9] = __builtin_amdgcn_mfma_f32_16x16x4f32(f3, f3, acc[9], 0, 0, 0);
    }
    als_lds_sync();  // tile consumed before next stage overwrites
  }
}

// ---- bf16 Gram accumulation (v_mfma_f32_16x16x32_bf16) ----
// A/B lane map of the 16x16x32 family: lane holds 8 contiguous-k
// elements, i (or j) = lane&15, k = (lane>>4)*8 + e (verified on
// hardware by tests/test_gpu_cf.py bf16-vs-f32 equivalence). One
// fragment per 16-dim block serves as A of tile-row t AND B of
// tile-col t, exactly like the f32 path's shared fragment.
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

// bf16-replica gathers, DWORD-shaped: each lane loads one dword = TWO
// adjacent dims of one staged edge row (sub-dword gathers measured 2.5x
// below dword gather throughput — BENCHLOG r2.4; this keeps the halved
// bytes AND full-rate loads). Lanes 0..31 cover row r's 32 dim-pairs,
// lanes 32..63 row r+1's; 16 passes stage a 32-edge tile. The pair is
// split into the transposed [dim][edge] bf16 panel with two b16 writes;
// rhs folds from the staged panel. Requires even K.
__device__ __forceinline__ void als_stage_tile_bf2(const CFAlsArgs& a,
                                                   uint32_t mycol, int rem,
                                                   int lane, float myw,
                                                   AlsGramLdsBf* lds,
                                                   uint32_t pair[16]) {
  const uint32_t* tab = (const uint32_t*)a.oldv_bf;
  int half = lane >> 5;        // which of the pass's two rows
  int m = lane & 31;           // dim pair index (dims 2m, 2m+1)
  int kp = a.K >> 1;
#pragma unroll
  for (int pass = 0; pass < 16; pass++) {
    int r = pass * 2 + half;
    uint32_t src = __shfl(mycol, r, WAVE);
    pair[pass] = (r < rem && m < kp)
                     ? tab[(uint64_t)src * kp + m]
                     : 0u;
  }
  if (lane < ALS_TILE) lds->W[lane] = myw;
#pragma unroll
  for (int pass = 0; pass < 16; pass++) {
    int r = pass * 2 + half;
    lds->S[(uint32_t)(2 * m) * ALS_BPITCH + r] =
        ((const __bf16*)&pair[pass])[0];
    lds->S[(uint32_t)(2 * m + 1) * ALS_BPITCH + r] =
        ((const __bf16*)&pair[pass])[1];
  }
}

__device__ __forceinline__ void als_gram_range_bf2(const CFAlsArgs& a,
                                                   E_ID b, E_ID e,
                                                   int lane,
                                                   AlsGramLdsBf* lds,
                                                   f32x4 acc[10],
                                                   float* rhs) {
  if (b >= e) return;
  int rem = (int)(e - b < ALS_TILE ? e - b : (E_ID)ALS_TILE);
  uint32_t mycol = 0;
  float myw = 0.0f;
  if (lane < rem) {
    mycol = a.col[b + lane];
    myw = (float)a.w[b + lane];
  }
  for (E_ID t = b; t < e; t += ALS_TILE) {
    uint32_t pair[16];
    als_stage_tile_bf2(a, mycol, rem, lane, myw, lds, pair);
    int rem_cur = rem;
    // prefetch next tile's col/weight
    E_ID t2 = t + ALS_TILE;
    bool more = t2 < e;
    uint32_t ncol = 0;
    float nw = 0.0f;
    if (more) {
      rem = (int)(e - t2 < ALS_TILE ? e - t2 : (E_ID)ALS_TILE);
      if (lane < rem) {
        ncol = a.col[t2 + lane];
        nw = (float)a.w[t2 + lane];
      }
    }
    als_lds_sync();
    // rhs += sum_r w_r * S[r][lane], from the staged bf16 panel
    if (lane < a.K) {
      const __bf16* row = &lds->S[(uint32_t)lane * ALS_BPITCH];
      float r0 = 0, r1 = 0;
      for (int r = 0; r + 2 <= rem_cur; r += 2) {
        r0 += lds->W[r] * (float)row[r];
        r1 += lds->W[r + 1] * (float)row[r + 1];
      }
      if (rem_cur & 1) r0 += lds->W[rem_cur - 1] * (float)row[rem_cur - 1];
      *rhs += r0 + r1;
    }
    bf16x8 f[4];
    int m = lane & 15, g = lane >> 4;
#pragma unroll
    for (int ti = 0; ti < 4; ti++)
      f[ti] = *(const bf16x8*)&lds->S[(uint32_t)(16 * ti + m) * ALS_BPITCH +
                                      g * 8];
    if (more) {  // next tile's gathers land under the MFMA burst... the
      myw = nw;  // bf2 path stages lazily per tile, so just roll state
      mycol = ncol;
    }
    acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[0], acc[0], 0,
                                                     0, 0);
    acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[1], acc[1], 0,
                                                     0, 0);
    acc[2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[2], acc[2], 0,
                                                     0, 0);
    acc[3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[3], acc[3], 0,
                                                     0, 0);
    acc[4] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[1], acc[4], 0,
                                                     0, 0);
    acc[5] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[2], acc[5], 0,
                                                     0, 0);
    acc[6] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[3], acc[6], 0,
                                                     0, 0);
    acc[7] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[2], f[2], acc[7], 0,
                                                     0, 0);
    acc[8] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[2], f[3], acc[8], 0,
                                                     0, 0);
    acc[9] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[3], f[3], acc[9], 0,
                                                     0, 0);
    als_lds_sync();
  }
}

__device__ __forceinline__ void als_gram_range_bf16(const CFAlsArgs& a,
                                                    E_ID b, E_ID e,
                                                    int lane,
                                                    AlsGramLdsBf* lds,
                                                    f32x4 acc[10],
                                                    float* rhs) {
  if (b >= e) return;
  int rem = (int)(e - b < ALS_TILE ? e - b : (E_ID)ALS_TILE);
  uint32_t mycol = 0;
  float myw = 0.0f;
  if (lane < rem) {
    mycol = a.col[b + lane];
    myw = (float)a.w[b + lane];
  }
  float tmp[ALS_TILE];
  als_vec_loads(a, mycol, rem, lane, tmp);
  for (E_ID t = b; t < e; t += ALS_TILE) {
    // rhs += sum_r w_r * S[r][lane] — straight from registers (lane=dim
    // holds S[r][lane] in tmp[r]); exact fp32
#pragma unroll
    for (int r = 0; r < ALS_TILE; r++)
      *rhs += __shfl(myw, r, WAVE) * tmp[r];
    // stage TRANSPOSED bf16: my row = dim `lane`, cols = edges (pairs
    // packed into dword stores)
    uint32_t* row32 = (uint32_t*)&lds->S[(uint32_t)lane * ALS_BPITCH];
#pragma unroll
    for (int w = 0; w < ALS_TILE / 2; w++) {
      union {
        __bf16 h[2];
        uint32_t u;
      } pk;
      pk.h[0] = (__bf16)tmp[2 * w];
      pk.h[1] = (__bf16)tmp[2 * w + 1];
      row32[w] = pk.u;
    }
    // prefetch next tile's col/weight, then its vectors (they land
    // under the MFMA burst)
    E_ID t2 = t + ALS_TILE;
    bool more = t2 < e;
    uint32_t ncol = 0;
    float nw = 0.0f;
    if (more) {
      rem = (int)(e - t2 < ALS_TILE ? e - t2 : (E_ID)ALS_TILE);
      if (lane < rem) {
        ncol = a.col[t2 + lane];
        nw = (float)a.w[t2 + lane];
      }
    }
    als_lds_sync();
    bf16x8 f[4];
    int m = lane & 15, g = lane >> 4;
#pragma unroll
    for (int ti = 0; ti < 4; ti++)
      f[ti] = *(const bf16x8*)&lds->S[(uint32_t)(16 * ti + m) * ALS_BPITCH +
                                      g * 8];
    if (more) {
      myw = nw;
      als_vec_loads(a, ncol, rem, lane, tmp);
      mycol = ncol;
    }
    acc[0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[0], acc[0], 0,
                                                     0, 0);
    acc[1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[1], acc[1], 0,
                                                     0, 0);
    acc[2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[2], acc[2], 0,
                                                     0, 0);
    acc[3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[0], f[3], acc[3], 0,
                                                     0, 0);
    acc[4] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[1], acc[4], 0,
                                                     0, 0);
    acc[5] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[2], acc[5], 0,
                                                     0, 0);
    acc[6] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[1], f[3], acc[6], 0,
                                                     0, 0);
    acc[7] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[2], f[2], acc[7], 0,
                                                     0, 0);
    acc[8] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[2], f[3], acc[8], 0,
                                                     0, 0);
    acc[9] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f[3], f[3], acc[9], 0,
                                                     0, 0);
    als_lds_sync();  // tile consumed before the next stage overwrites
  }
}

// Scatter MFMA accumulators into the LDS Gram (mirroring the lower half)
// using the 16x16x4 C/D lane map: element (row=(lane>>4)*4+reg, col=lane&15).
__device__ __forceinline__ void als_dump_gram(AlsLds* lds, f32x4 acc[10],
                                              int lane, int K) {
  static constexpr int TI[10] = {0, 0, 0, 0, 1, 1, 1, 2, 2, 3};
  static constexpr int TJ[10] = {0, 1, 2, 3, 1, 2, 3, 2, 3, 3};
  int row = (lane >> 4) * 4, col = lane & 15;
#pragma unroll
  for (int tidx = 0; tidx < 10; tidx++) {
#pragma unroll
    for (int reg = 0; reg < 4; reg++) {
      int gr = TI[tidx] * 16 + row + reg;
      int gc = TJ[tidx] * 16 + col;
      float v = acc[tidx][reg];
      lds->G[gr * ALS_ROW + gc] = v;
      if (TI[tidx] != TJ[tidx]) lds->G[gc * ALS_ROW + gr] = v;
    }
  }
  als_lds_sync();
  // regularizer + identity padding for unused dims (keeps G SPD for K<64)
  lds->G[lane * ALS_ROW + lane] =
      lane < K ? lds->G[lane * ALS_ROW + lane] + CF_LAMBDA : 1.0f;
  als_lds_sync();
}

// In-place lower Cholesky of the 64x64 LDS matrix (one wave; lane = row).
// Semi-definite handling: a (near-)rank-deficient Gram — e.g. the parity
// init makes sweep 1's Gram rank-1 with entries ~deg/K, whose
// elimination cancels later pivots to rounding noise (negative under
// bf16 inputs at NetFlix-scale hub degrees; sqrt -> NaN, and a naive
// lambda floor instead explodes the degenerate directions' step, which
// overflowed the NEXT sweep's Gram to inf/NaN). Deficient pivots are
// marked with a -1 sentinel, their column is zeroed, and the solve
// takes NO step along those directions — the exact-arithmetic
// pseudoinverse behaviour on the deficient subspace.
__device__ __forceinline__ void wave_cholesky64(float* G, int lane) {
  for (int k = 0; k < ALS_K; k++) {
    float piv = G[k * ALS_ROW + k];
    bool ok = piv > (float)CF_LAMBDA * 0.5f;
    float dkk = ok ? sqrtf(piv) : 1.0f;
    float lik = (lane > k && ok) ? G[lane * ALS_ROW + k] / dkk : 0.0f;
    if (lane == k) G[k * ALS_ROW + k] = ok ? dkk : -1.0f;
    if (lane > k) G[lane * ALS_ROW + k] = lik;
    als_lds_sync();
    for (int j = k + 1; j <= lane; j++)
      G[lane * ALS_ROW + j] -= lik * G[j * ALS_ROW + k];
    als_lds_sync();
  }
}

// Solve L L^T d = rhs; rhs/result live lane=dim in a register.
// Sentinel (-1) pivots contribute zero (see wave_cholesky64).
__device__ __forceinline__ float wave_spd_solve64(const float* G, float r,
                                                  int lane) {
  for (int k = 0; k < ALS_K; k++) {  // forward: L y = r
    float lkk = G[k * ALS_ROW + k];
    float yk = lkk > 0.0f ? __shfl(r, k, WAVE) / lkk : 0.0f;
    if (lane == k) r = yk;
    else if (lane > k) r -= G[lane * ALS_ROW + k] * yk;
  }
  for (int k = ALS_K - 1; k >= 0; k--) {  // backward: L^T d = y
    float lkk = G[k * ALS_ROW + k];
    float dk = lkk > 0.0f ? __shfl(r, k, WAVE) / lkk : 0.0f;
    if (lane == k) r = dk;
    else if (lane < k) r -= G[k * ALS_ROW + lane] * dk;
  }
  return r;
}

// ---- one wave per vertex: gram + solve fused (deg < T2 bin lists) ----
// MODE selects the Gram path at compile time (0 = exact fp32, 1 = bf16
// MFMA with fp32 gathers, 2 = bf16 MFMA with dword-pair bf16 gathers):
// a runtime branch would co-inline every path and the register
// allocator sizes for their union (measured: +40 VGPRs, occupancy 2->1,
// 10.5 -> 18 ms/sweep with the branch never taken).
template <int MODE>
__global__ __launch_bounds__(ALS_TB) void cf_als_solve_kernel(
    uint32_t n, const V_ID* binlist, CFAlsArgs a) {
  __shared__ AlsLds lds;
  int lane = threadIdx.x;
  for (uint64_t i = blockIdx.x; i < n; i += gridDim.x) {
    V_ID v = binlist[i];
    E_ID b = a.row_ptr[v], e = a.row_ptr[v + 1];
    f32x4 acc[10];
#pragma unroll
    for (int t = 0; t < 10; t++) acc[t] = {0, 0, 0, 0};
    float rhs = 0.0f;
    if (MODE == 2)
      als_gram_range_bf2(a, b, e, lane, &lds.gb, acc, &rhs);
    else if (MODE == 1)
      als_gram_range_bf16(a, b, e, lane, &lds.gb, acc, &rhs);
    else
      als_gram_range(a, b, e, lane, &lds.g, acc, &rhs);
    als_dump_gram(&lds, acc, lane, a.K);
    wave_cholesky64(lds.G, lane);
    float d = wave_spd_solve64(lds.G, rhs, lane);
    if (lane < a.K) a.newv[(uint64_t)v * a.K + lane] = d;
    als_lds_sync();  // G reads done before next vertex's MFMA dump
  }
}

// ---- hub path: chunk-parallel Gram into global scratch ----
// gram_scratch: f32[nbig * 64 * 64] (upper-triangular elements only),
// rhs_scratch: f32[nbig * 64]; both pre-zeroed by the engine each sweep.
constexpr int ALS_CHUNK_TB = 128;  // 2 independent waves per workgroup

template <int MODE>
__global__ __launch_bounds__(ALS_CHUNK_TB) void cf_als_gram_chunk_kernel(
    uint32_t n2, const uint2* bin2, V_ID chunk_edges, const int* hubidx,
    float* gram_scratch, float* rhs_scratch, CFAlsArgs a) {
  union StageLds {
    AlsGramLds g;
    AlsGramLdsBf gb;
  };
  __shared__ StageLds lds2[ALS_CHUNK_TB / WAVE];
  StageLds& lds = lds2[threadIdx.x >> 6];
  int lane = threadIdx.x & (WAVE - 1);
  uint32_t wave = (blockIdx.x * (ALS_CHUNK_TB / WAVE)) + (threadIdx.x >> 6);
  uint32_t nwaves = gridDim.x * (ALS_CHUNK_TB / WAVE);
  for (uint32_t i = wave; i < n2; i += nwaves) {
    uint2 ent = bin2[i];
    V_ID v = ent.x;
    E_ID b = a.row_ptr[v] + (E_ID)ent.y * chunk_edges;
    E_ID e = a.row_ptr[v + 1];
    if (e > b + chunk_edges) e = b + chunk_edges;
    f32x4 acc[10];
#pragma unroll
    for (int t = 0; t < 10; t++) acc[t] = {0, 0, 0, 0};
    float rhs = 0.0f;
    if (MODE == 2)
      als_gram_range_bf2(a, b, e, lane, &lds.gb, acc, &rhs);
    else if (MODE == 1)
      als_gram_range_bf16(a, b, e, lane, &lds.gb, acc, &rhs);
    else
      als_gram_range(a, b, e, lane, &lds.g, acc, &rhs);
    int idx = hubidx[v];
    float* Gg = gram_scratch + (uint64_t)idx * ALS_K * ALS_K;
    atomicAdd(&rhs_scratch[(uint64_t)idx * ALS_K + lane], rhs);
    static constexpr int TI[10] = {0, 0, 0, 0, 1, 1, 1, 2, 2, 3};
    static constexpr int TJ[10] = {0, 1, 2, 3, 1, 2, 3, 2, 3, 3};
    int row = (lane >> 4) * 4, col = lane & 15;
#pragma unroll
    for (int tidx = 0; tidx < 10; tidx++) {
#pragma unroll
      for (int reg = 0; reg < 4; reg++) {
        int gr = TI[tidx] * 16 + row + reg;
        int gc = TJ[tidx] * 16 + col;
        if (gc >= gr) atomicAdd(&Gg[gr * ALS_K + gc], acc[tidx][reg]);
      }
    }
  }
}

// ---- hub path: per-vertex solve from global scratch ----
__global__ __launch_bounds__(ALS_TB) void cf_als_hub_solve_kernel(
    uint32_t nbig, const V_ID* bin2v, const float* gram_scratch,
    const float* rhs_scratch, CFAlsArgs a) {
  __shared__ AlsLds lds;
  int lane = threadIdx.x;
  for (uint32_t i = blockIdx.x; i < nbig; i += gridDim.x) {
    V_ID v = bin2v[i];
    const float* Gg = gram_scratch + (uint64_t)i * ALS_K * ALS_K;
    for (int r = 0; r < ALS_K; r++) {
      float x = Gg[r * ALS_K + lane];
      lds.G[r * ALS_ROW + lane] = x;  // upper half valid (lane >= r)
    }
    als_lds_sync();
    for (int r = 0; r < ALS_K; r++)  // mirror upper -> lower
      if (lane > r) lds.G[lane * ALS_ROW + r] = lds.G[r * ALS_ROW + lane];
    als_lds_sync();
    lds.G[lane * ALS_ROW + lane] =
        lane < a.K ? lds.G[lane * ALS_ROW + lane] + CF_LAMBDA : 1.0f;
    float rhs = rhs_scratch[(uint64_t)i * ALS_K + lane];
    als_lds_sync();
    wave_cholesky64(lds.G, lane);
    float d = wave_spd_solve64(lds.G, rhs, lane);
    if (lane < a.K) a.newv[(uint64_t)v * a.K + lane] = d;
    als_lds_sync();
  }
}

}  // namespace lux

using namespace lux;

extern "C" {

// One ALS sweep over my partition. Reuses the pull.hip degree-bin lists;
// hubidx maps a hub vertex (local id) to its scratch slot [0, nbig).
// gram_scratch/rhs_scratch must be zeroed before each sweep when nbig > 0.
void lux_gpu_cf_als_iter(uint64_t stream, uint32_t n0, const V_ID* bin0,
                         uint32_t n1, const V_ID* bin1, uint32_t n2,
                         const uint2* bin2, uint32_t nbig, const V_ID* bin2v,
                         const int* hubidx, float* gram_scratch,
                         float* rhs_scratch, const E_ID* row_ptr,
                         const V_ID* col, const WeightType* w,
                         const float* oldv,
                         const uint16_t* oldv_bf /*nullable*/, float* newv,
                         V_ID row_left, int K) {
  hipStream_t s = (hipStream_t)stream;
  CFAlsArgs a{row_ptr, col, w, oldv, (const __bf16*)oldv_bf, newv, row_left,
              K};
  // Mode defaults are PER KERNEL CLASS. Fused per-row solves: exact fp32
  // Gram (16x16x4). bf16 Gram noise (~0.4% of ||G||) swamps sigma_min for
  // moderate-degree rows during alternation — measured: item factors jump
  // 0.32 -> 409 after one bf16 half-sweep and the feedback diverges to
  // 1e13 loss. Chunk/hub Gram (deg >= T2 = 2048): bf16 16x16x32 MFMA
  // stays the default — averaging over >= 2048 sources keeps ||G|| /
  // sigma_min well above the bf16 noise floor, and that is where the
  // NetFlix-shaped item volume (avg item degree ~5.7K) lives.
  // LUX_ALS_F32=1 forces fp32 everywhere; LUX_ALS_BF16=1 forces bf16 for
  // the fused solves too; mode 2 = dword-pair bf16 gathers (only when
  // the engine passed a bf16 replica; LUX_ALS_BF_GATHER experiment).
  int mode = getenv("LUX_ALS_F32")
                 ? 0
                 : (oldv_bf ? 2 : (getenv("LUX_ALS_BF16") ? 1 : 0));
  int chunk_mode = getenv("LUX_ALS_F32") ? 0 : (oldv_bf ? 2 : 1);
#define LUX_ALS_LAUNCH(K_, GRID_, N_, LIST_)                                  do {                                                                          if (mode == 2)                                                                hipLaunchKernelGGL(K_<2>, GRID_, dim3(ALS_TB), 0, s, N_, LIST_, a);       else if (mode == 1)                                                           hipLaunchKernelGGL(K_<1>, GRID_, dim3(ALS_TB), 0, s, N_, LIST_, a);       else                                                                          hipLaunchKernelGGL(K_<0>, GRID_, dim3(ALS_TB), 0, s, N_, LIST_, a);     } while (0)
  if (nbig) {
    uint32_t gw = (n2 + 1) / 2;
    dim3 grid(gw > MAX_GRID ? MAX_GRID : gw);
    int mode = chunk_mode;  // hub Gram keeps bf16 (see above)
    if (mode == 2)
      hipLaunchKernelGGL(cf_als_gram_chunk_kernel<2>, grid,
                         dim3(ALS_CHUNK_TB), 0, s, n2, bin2, (V_ID)8192,
                         hubidx, gram_scratch, rhs_scratch, a);
    else if (mode == 1)
      hipLaunchKernelGGL(cf_als_gram_chunk_kernel<1>, grid,
                         dim3(ALS_CHUNK_TB), 0, s, n2, bin2, (V_ID)8192,
                         hubidx, gram_scratch, rhs_scratch, a);
    else
      hipLaunchKernelGGL(cf_als_gram_chunk_kernel<0>, grid,
                         dim3(ALS_CHUNK_TB), 0, s, n2, bin2, (V_ID)8192,
                         hubidx, gram_scratch, rhs_scratch, a);
    hipLaunchKernelGGL(cf_als_hub_solve_kernel,
                       dim3(nbig > MAX_GRID ? MAX_GRID : nbig), dim3(ALS_TB),
                       0, s, nbig, bin2v, gram_scratch, rhs_scratch, a);
  }
  if (n1)
    LUX_ALS_LAUNCH(cf_als_solve_kernel, dim3(n1 > MAX_GRID ? MAX_GRID : n1),
                   n1, bin1);
  if (n0)
    LUX_ALS_LAUNCH(cf_als_solve_kernel, dim3(n0 > MAX_GRID ? MAX_GRID : n0),
                   n0, bin0);
#undef LUX_ALS_LAUNCH
  LUX_POST_LAUNCH(stream);
}

}  // extern "C"
