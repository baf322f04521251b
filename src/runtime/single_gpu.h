// Native single-GPU runtime: FB arena allocator + engine drivers that run
// the gfx950 kernels without Python. Used by the CLI app binaries
// (apps/*.cpp) — the MI355X equivalents of the reference's per-app
// top_level_task drivers (pagerank/pagerank.cc:32-119 etc.), with the
// Legion region machinery replaced by explicit device buffers on one HIP
// stream. (Multi-GPU runs go through the torch.distributed/RCCL Python
// engine — one process per GPU.)
#pragma once
#include <hip/hip_runtime.h>

#include <cstdint>
#include <vector>

#include "lux/graph.h"
#include "lux/gpu_api.h"
#include "lux/types.h"

#define LUX_OK(cmd)                                                         \
  do {                                                                      \
    hipError_t e_ = (cmd);                                                  \
    if (e_ != hipSuccess) {                                                 \
      fprintf(stderr, "HIP error %s:%d: %s\n", __FILE__, __LINE__,          \
              hipGetErrorString(e_));                                       \
      abort();                                                              \
    }                                                                       \
  } while (0)

namespace lux {

// FB memory pool: one upfront hipMalloc, bump allocation, 256-B alignment
// (we own our arena — no Realm-internal back-doors, SURVEY.md §7 "fix").
class DeviceArena {
 public:
  explicit DeviceArena(size_t bytes);
  ~DeviceArena();
  void* alloc(size_t bytes);
  template <typename T>
  T* alloc_n(size_t n) {
    return (T*)alloc(n * sizeof(T));
  }
  size_t used() const { return used_; }
  size_t capacity() const { return cap_; }

 private:
  char* base_ = nullptr;
  size_t cap_ = 0, used_ = 0;
};

struct DeviceGraph {
  V_ID nv = 0;
  E_ID ne = 0;
  E_ID* col_end = nullptr;   // u64[nv] end offsets
  V_ID* src = nullptr;       // u32[ne]
  WeightType* weight = nullptr;

  static DeviceGraph upload(const HostCSC& g, DeviceArena& arena,
                            hipStream_t s);
  // Synthetic generation fully on-device.
  static DeviceGraph rmat(int scale, E_ID ne, uint64_t seed,
                          DeviceArena& arena, hipStream_t s);
  static DeviceGraph bipartite(V_ID n_users, V_ID n_items, E_ID ne,
                               uint64_t seed, DeviceArena& arena,
                               hipStream_t s);
};

// Degree bins shared by the pull/CF engines (built once; see pull.hip).
struct Bins {
  V_ID *bin0, *bin1, *bin2v;
  lux_uint2* bin2;
  uint32_t n0, n1, n2, nbig;
  void build(const E_ID* row_ptr_loc, V_ID vp, E_ID ep, DeviceArena& arena,
             hipStream_t s);
};

// Stable-partition counts after splitting every bin list at a local row
// boundary (rows < lb first). Used by ALS alternation: the user
// half-sweep runs the first n*u entries of each list, the item half-sweep
// the rest via offset pointers. Init-time host roundtrip.
struct BinSplit {
  uint32_t n0u = 0, n1u = 0, n2u = 0, nbigu = 0;
};
BinSplit split_bins_at(Bins& b, V_ID lb, hipStream_t s);

// Deterministic jittered ALS init: sqrt(1/K) * [0.5, 1.5) per component
// (splitmix64 of the flat index; identical on every rank — matches
// lux_amd/cf_engine.py als_init). The constant sqrt(1/K) init makes the
// first alternating half-sweep rank-1 degenerate (all users land on one
// line; the item solve amplifies null-space noise by 1/lambda and the
// bf16 path diverges), so the exact-solve optimizer breaks the
// degeneracy at init. SGD keeps the reference's constant init.
inline float als_init_val(uint64_t i, int K) {
  uint64_t z = (i + 0x9E3779B97F4A7C15ull) * 0xBF58476D1CE4E5B9ull;
  z ^= z >> 30;
  z *= 0x94D049BB133111EBull;
  z ^= z >> 27;
  double u = (double)(z >> 11) * (1.0 / 9007199254740992.0);
  return (float)(sqrt(1.0 / K) * (0.5 + u));
}

// src-blocked CSC + per-block compacted bins — the native twin of
// engine.py build_blocked: regroups edges by 32 MB src window so the
// random old-property gather stays Infinity-Cache-resident, with
// u32 block-local row offsets (half the per-row sweep traffic).
// Built only when the gather window exceeds the LLC (nv*4 B > 256 MiB).
struct BlockedPull {
  struct Blk {
    uint32_t* row32;
    V_ID* col;
    uint32_t n0, n1, n2, nbig;
    V_ID *bin0, *bin1, *bin2v;
    lux_uint2* bin2;
  };
  std::vector<Blk> blocks;
  V_ID vp = 0;
  bool active() const { return !blocks.empty(); }
  static size_t arena_bytes(V_ID nv, V_ID vp, E_ID ep);
  void build(const E_ID* row_ptr_loc, const V_ID* col, V_ID vp, E_ID ep,
             V_ID nv, DeviceArena& arena, hipStream_t s);
  // one fold-sweep over every block (same iteration contract as the
  // unblocked pull: newv pre-seeded, PR epilogue applied by the caller)
  void sweep(int mode, const void* oldv, void* newv, const V_ID* deg,
             V_ID row_left, float init_rank, hipStream_t s) const;
};

class SingleGpuPagerank {
 public:
  SingleGpuPagerank(const DeviceGraph& g, DeviceArena& arena, hipStream_t s);
  void iterate(int iters);
  const float* ranks() const { return old_; }  // device ptr, stored pr/deg

 private:
  const DeviceGraph& g_;
  hipStream_t s_;
  E_ID* row_ptr_;
  Bins bins_;
  BlockedPull blocked_;
  V_ID* deg_;
  float *old_, *new_;
};

class SingleGpuPush {
 public:
  SingleGpuPush(const DeviceGraph& g, bool is_min, V_ID source,
                DeviceArena& arena, hipStream_t s, bool verbose = false);
  // returns iterations to convergence
  int run(int max_iters = 0);
  uint64_t check();  // violation count (check oracle)
  const uint32_t* labels() const { return labels_; }

 private:
  V_ID step();  // returns new frontier size
  const DeviceGraph& g_;
  hipStream_t s_;
  bool is_min_, verbose_;
  E_ID *row_ptr_, *push_row_ptr_;
  V_ID* push_col_;
  Bins bins_;
  BlockedPull blocked_;
  uint32_t* bits_ = nullptr;  // BFS visited bitmap (is_min fast path)
  bool bits_stale_ = true;
  uint32_t *labels_, *snapshot_;
  uint8_t *fq_, *new_fq_, *tmp_fq_;
  lux_uint2* items_;
  uint32_t* item_counter_;
  uint32_t max_items_;
  V_ID capacity_;
  uint32_t fq_type_, fq_num_;
  bool force_pull_ = false;  // set on expand overflow (loud recovery)
  int iters_ = 0;
};

// Union-find CC (cc_uf.hip): one edge pass + Afforest sampling; identical
// labelling to converged max-label propagation on symmetric inputs
// (components -labelprop selects the reference-parity SingleGpuPush).
class SingleGpuCCUnionFind {
 public:
  SingleGpuCCUnionFind(const DeviceGraph& g, DeviceArena& arena,
                       hipStream_t s);
  void run();
  uint64_t check();
  const uint32_t* labels() const { return (const uint32_t*)labels_; }

 private:
  const DeviceGraph& g_;
  hipStream_t s_;
  E_ID* row_ptr_;
  Bins bins_;
  V_ID *parent_, *labels_;
  uint32_t* gbits_;
};

class SingleGpuCF {
 public:
  // als: exact MFMA ALS sweeps (cf_als.hip, K <= 64) instead of SGD.
  // n_users > 0 (bipartite boundary) turns each ALS sweep into true
  // Gauss-Seidel alternation: user rows solve against old item factors,
  // item rows against the UPDATED users (exact in place — a bipartite
  // row only reads the other side). 0 = unknown -> simultaneous Jacobi.
  SingleGpuCF(const DeviceGraph& g, int K, DeviceArena& arena,
              hipStream_t s, bool als = false, V_ID n_users = 0);
  void iterate(int iters);
  const float* vectors() const { return old_; }

 private:
  const DeviceGraph& g_;
  hipStream_t s_;
  int K_;
  bool als_;
  E_ID* row_ptr_;
  Bins bins_;
  float *old_, *new_;
  int* hubidx_ = nullptr;       // ALS hub scratch slot map
  float* gram_ = nullptr;       // ALS: nbig x 64 x 64
  float* rhs_ = nullptr;        // ALS: nbig x 64
  V_ID nu_ = 0;                 // bipartite boundary (0 = Jacobi)
  BinSplit split_;              // bin counts below nu_ (alternation)
};

}  // namespace lux
