// Core types shared by every layer of the MI355X-native Lux rebuild.
//
// Semantics mirror the reference's app.h type choices
// (/root/reference/pagerank/app.h:21-36, /root/reference/sssp/app.h:19-39,
// /root/reference/col_filter/app.h:20-43) but are owned here as one header:
// V_ID (vertex id) is u32, E_ID (edge offset) is u64, edge weights are i32,
// PageRank/CF state is f32, SSSP/CC labels are u32.
#pragma once
#include <cstdint>

namespace lux {

using V_ID = uint32_t;
using E_ID = uint64_t;
using WeightType = int32_t;

// SSSP/CC label "infinity" (sssp initialises every non-source label to the
// max representable value; reference seeds via memset-like CPU loop).
constexpr V_ID INF_LABEL = 0xFFFFFFFFu;

// PageRank constants (reference pagerank/app.h:24 uses ALPHA exactly like
// this: new_pr = (1-ALPHA)/nv + ALPHA * sum_in(old_pr_scaled)).
constexpr float PR_ALPHA = 0.15f;

// Collaborative filtering constants (reference col_filter/app.h:26-28).
// K (the latent rank) is a runtime parameter here (reference hardcodes 20;
// our benchmark config uses 64).
constexpr float CF_LAMBDA = 0.001f;
constexpr float CF_GAMMA = 0.00000035f;

// Frontier segment header, byte-compatible with the reference's
// FrontierHeader (core/graph.h:100-106): a type tag followed by the number
// of active vertices in this segment.
struct FrontierHeader {
  static constexpr uint32_t DENSE_BITMAP = 0x1234567u;
  static constexpr uint32_t SPARSE_QUEUE = 0x7654321u;
  uint32_t type;
  V_ID numNodes;
};
static_assert(sizeof(FrontierHeader) == 8, "FrontierHeader must be 8 bytes");

// Frontier sizing rule (reference core/push_model.inl:391-412):
// a partition with `verts` vertices gets a frontier byte range of
// (verts/SPARSE_THRESHOLD + 100) * sizeof(V_ID) + sizeof(FrontierHeader),
// and the push model falls back to a dense pull iteration when the total
// frontier exceeds nv / SPARSE_THRESHOLD (sssp_gpu.cu:414).
constexpr V_ID SPARSE_THRESHOLD = 16;
constexpr int SLIDING_WINDOW = 4;

inline uint64_t frontier_bytes(V_ID part_verts) {
  return (uint64_t(part_verts) / SPARSE_THRESHOLD + 100) * sizeof(V_ID) +
         sizeof(FrontierHeader);
}
inline V_ID frontier_capacity(V_ID part_verts) {
  // max sparse-queue entries in a segment (sssp_gpu.cu:410).
  return part_verts / SPARSE_THRESHOLD + 100;
}

}  // namespace lux
