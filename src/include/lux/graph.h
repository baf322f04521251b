// Host-side CSC graph container + .lux binary IO + edge-balanced partitioner.
//
// The .lux on-disk format is byte-compatible with the reference
// (README.md:56-75, tools/converter.cc:108-124):
//   u32 nv; u64 ne; u64 col_end[nv]; u32 src[ne]; [i32 weight[ne]]
// col_end[i] is the END offset of vertex i's in-edge block (col_end[-1]=0
// implied). A trailing u32 degree[nv] block written by the reference
// converter is tolerated and ignored on read (no reference loader reads it;
// see core/pull_model.inl:253-320).
#pragma once
#include <cstdio>
#include <string>
#include <vector>

#include "types.h"

namespace lux {

struct HostCSC {
  V_ID nv = 0;
  E_ID ne = 0;
  std::vector<E_ID> col_end;        // size nv (end offsets, .lux convention)
  std::vector<V_ID> src;            // size ne, grouped by dst
  std::vector<WeightType> weight;   // size ne if weighted, else empty
  bool weighted() const { return !weight.empty(); }

  E_ID row_begin(V_ID v) const { return v == 0 ? 0 : col_end[v - 1]; }
  E_ID row_end(V_ID v) const { return col_end[v]; }
};

// ---- IO ----
bool lux_write(const std::string& path, const HostCSC& g);
bool lux_read_header(const std::string& path, V_ID* nv, E_ID* ne,
                     bool* weighted);
bool lux_read(const std::string& path, HostCSC* g, bool want_weights);
// Partition-slice read: reads only [row_left, row_right] rows' offsets and
// their edge block (the MI355X equivalent of the per-partition
// fseeko/fread load task, core/pull_model.inl:253-320).
bool lux_read_slice(const std::string& path, V_ID row_left, V_ID row_right,
                    std::vector<E_ID>* col_end, std::vector<V_ID>* src,
                    std::vector<WeightType>* weight);

// ---- Synthetic generation (CPU) ----
HostCSC rmat_csc_cpu(int scale, E_ID ne, uint64_t seed);
HostCSC bipartite_csc_cpu(V_ID n_users, V_ID n_items, E_ID ne, uint64_t seed);
// Build CSC from an explicit edge list (counting sort by dst, the same
// algorithm the GPU builder uses).
HostCSC edges_to_csc(V_ID nv, const std::vector<V_ID>& src,
                     const std::vector<V_ID>& dst,
                     const std::vector<WeightType>* w);

// ---- Partitioner ----
// Edge-balanced contiguous vertex ranges, one per part (the reference's
// scheme: edge_cap = ceil(ne/p), greedy cut — core/pull_model.inl:108-131).
// Returns row_left/row_right inclusive bounds per part; parts may be empty
// (row_left > row_right) when nv < nparts.
struct Partition {
  std::vector<V_ID> row_left, row_right;  // size nparts, inclusive
  std::vector<E_ID> col_left, col_right;  // edge ranges [col_left, col_right)
};
Partition partition_edge_balanced(V_ID nv, E_ID ne, const E_ID* col_end,
                                  int nparts);

}  // namespace lux
