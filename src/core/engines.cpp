// CPU reference engines for the four vertex programs.
//
// These are the golden-semantics implementations the GPU kernels are tested
// against, and the per-partition *_iter_part entry points are
// the compute step of the CPU multi-process path (gloo, world_size>1) that
// exercises the distributed exchange layer without a GPU.
//
// Semantics parity (reference file:line):
//  - PageRank: new_pr = (1-ALPHA)/nv + ALPHA * sum_{u in in(v)} old_pr[u],
//    stored divided by out-degree (pagerank_gpu.cu:97-100); initial value
//    rank/degree with rank = 1/nv, degree-0 stores rank (pagerank_gpu.cu:255-259).
//  - SSSP: unweighted hop relaxation label[dst] = min(label[dst],
//    label[src]+1) (sssp_gpu.cu:122,208,225); source = 0, rest INF.
//  - CC: label[dst] = max(label[dst], label[src]) along directed edges
//    (components_gpu.cu:122); init label[v] = v.
//  - CF: one SGD sweep per iteration over in-edges of each dst using OLD
//    vectors throughout: err = w - <src_vec, dst_vec>; acc += err*src_vec;
//    new = old + GAMMA*(acc - LAMBDA*old) (colfilter_gpu.cu:83-101). The
//    reference indexes the dst vector with a partition-local offset into a
//    full-length array (colfilter_gpu.cu:65 — a rowLeft>0 bug); we use the
//    intended old[dst].
#include <algorithm>
#include <cmath>
#include <cstring>
#include <vector>

#include "lux/graph.h"

namespace lux {

// ---------- PageRank (pull) ----------

void pagerank_init(V_ID nv, const V_ID* out_deg, float* pr) {
  float rank = 1.0f / nv;
  for (V_ID v = 0; v < nv; v++)
    pr[v] = out_deg[v] == 0 ? rank : rank / out_deg[v];
}

// One iteration over partition rows [row_left, row_right]; col_end/src are
// the partition slice (col_end global offsets, src indexed from col_left).
void pagerank_iter_part(V_ID nv, V_ID row_left, V_ID row_right,
                        E_ID col_left, const E_ID* col_end, const V_ID* src,
                        const V_ID* out_deg, const float* old_pr,
                        float* new_pr_part) {
  float init_rank = (1.0f - PR_ALPHA) / nv;
  for (V_ID v = row_left; v <= row_right; v++) {
    E_ID b = (v == row_left) ? col_left : col_end[v - 1 - row_left];
    E_ID e = col_end[v - row_left];
    float sum = 0.0f;
    for (E_ID i = b; i < e; i++) sum += old_pr[src[i - col_left]];
    float pr = init_rank + PR_ALPHA * sum;
    new_pr_part[v - row_left] = out_deg[v] == 0 ? pr : pr / out_deg[v];
  }
}

void out_degrees(V_ID nv, E_ID ne, const V_ID* src, V_ID* deg) {
  std::memset(deg, 0, sizeof(V_ID) * nv);
  for (E_ID e = 0; e < ne; e++) deg[src[e]]++;
}

// Whole-graph driver (tests/config-1 plumbing path).
void pagerank_cpu(const HostCSC& g, int iters, float* pr_out) {
  std::vector<V_ID> deg(g.nv);
  out_degrees(g.nv, g.ne, g.src.data(), deg.data());
  std::vector<float> oldp(g.nv), newp(g.nv);
  pagerank_init(g.nv, deg.data(), oldp.data());
  for (int it = 0; it < iters; it++) {
    pagerank_iter_part(g.nv, 0, g.nv - 1, 0, g.col_end.data(), g.src.data(),
                       deg.data(), oldp.data(), newp.data());
    std::swap(oldp, newp);
  }
  std::memcpy(pr_out, oldp.data(), sizeof(float) * g.nv);
}

// ---------- Label propagation core (SSSP min / CC max) ----------

// Dense pull relaxation of one partition: for every dst row, fold in-edge
// source labels. Returns number of labels that changed.
template <bool IS_MIN>
static V_ID label_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                            const E_ID* col_end, const V_ID* src,
                            const V_ID* old_label, V_ID* new_label_part) {
  V_ID changed = 0;
  for (V_ID v = row_left; v <= row_right; v++) {
    E_ID b = (v == row_left) ? col_left : col_end[v - 1 - row_left];
    E_ID e = col_end[v - row_left];
    V_ID lab = old_label[v];
    for (E_ID i = b; i < e; i++) {
      V_ID sl = old_label[src[i - col_left]];
      if (IS_MIN) {
        V_ID cand = sl == INF_LABEL ? INF_LABEL : sl + 1;
        lab = std::min(lab, cand);
      } else {
        lab = std::max(lab, sl);
      }
    }
    if (lab != old_label[v]) changed++;
    new_label_part[v - row_left] = lab;
  }
  return changed;
}

V_ID sssp_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                    const E_ID* col_end, const V_ID* src,
                    const V_ID* old_label, V_ID* new_label_part) {
  return label_iter_part<true>(row_left, row_right, col_left, col_end, src,
                               old_label, new_label_part);
}
V_ID cc_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                  const E_ID* col_end, const V_ID* src, const V_ID* old_label,
                  V_ID* new_label_part) {
  return label_iter_part<false>(row_left, row_right, col_left, col_end, src,
                                old_label, new_label_part);
}

// Fixed-point drivers (converged results == what the push engine converges
// to; iteration count bounded by nv).
int sssp_cpu(const HostCSC& g, V_ID source, V_ID* label_out) {
  std::vector<V_ID> oldl(g.nv, INF_LABEL), newl(g.nv);
  oldl[source] = 0;
  int iters = 0;
  while (true) {
    V_ID changed = sssp_iter_part(0, g.nv - 1, 0, g.col_end.data(),
                                  g.src.data(), oldl.data(), newl.data());
    iters++;
    std::swap(oldl, newl);
    if (changed == 0) break;
  }
  std::memcpy(label_out, oldl.data(), sizeof(V_ID) * g.nv);
  return iters;
}

int cc_cpu(const HostCSC& g, V_ID* label_out) {
  std::vector<V_ID> oldl(g.nv), newl(g.nv);
  for (V_ID v = 0; v < g.nv; v++) oldl[v] = v;
  int iters = 0;
  while (true) {
    V_ID changed = cc_iter_part(0, g.nv - 1, 0, g.col_end.data(),
                                g.src.data(), oldl.data(), newl.data());
    iters++;
    std::swap(oldl, newl);
    if (changed == 0) break;
  }
  std::memcpy(label_out, oldl.data(), sizeof(V_ID) * g.nv);
  return iters;
}

// Check oracles, per the reference's CheckTask kernels
// (sssp_gpu.cu:773-798, components_gpu.cu:767-791): count violating edges.
E_ID sssp_check(const HostCSC& g, const V_ID* label) {
  E_ID mistakes = 0;
  for (V_ID v = 0; v < g.nv; v++) {
    for (E_ID i = g.row_begin(v); i < g.row_end(v); i++) {
      V_ID sl = label[g.src[i]];
      V_ID cand = sl == INF_LABEL ? INF_LABEL : sl + 1;
      if (label[v] > cand) mistakes++;
    }
  }
  return mistakes;
}
E_ID cc_check(const HostCSC& g, const V_ID* label) {
  E_ID mistakes = 0;
  for (V_ID v = 0; v < g.nv; v++)
    for (E_ID i = g.row_begin(v); i < g.row_end(v); i++)
      if (label[v] < label[g.src[i]]) mistakes++;
  return mistakes;
}

// ---------- Collaborative filtering ----------

void cf_init(V_ID nv, int K, float* vec) {
  float v0 = std::sqrt(1.0f / K);  // reference init (colfilter_gpu.cu:260-264)
  for (uint64_t i = 0; i < (uint64_t)nv * K; i++) vec[i] = v0;
}

void cf_iter_part(V_ID row_left, V_ID row_right, E_ID col_left,
                  const E_ID* col_end, const V_ID* src, const WeightType* w,
                  int K, const float* old_vec, float* new_vec_part) {
  std::vector<float> acc(K);
  for (V_ID v = row_left; v <= row_right; v++) {
    E_ID b = (v == row_left) ? col_left : col_end[v - 1 - row_left];
    E_ID e = col_end[v - row_left];
    const float* dv = old_vec + (uint64_t)v * K;
    std::fill(acc.begin(), acc.end(), 0.0f);
    for (E_ID i = b; i < e; i++) {
      const float* sv = old_vec + (uint64_t)src[i - col_left] * K;
      float dot = 0.0f;
      for (int k = 0; k < K; k++) dot += sv[k] * dv[k];
      float err = (float)w[i - col_left] - dot;
      for (int k = 0; k < K; k++) acc[k] += err * sv[k];
    }
    float* nv_ = new_vec_part + (uint64_t)(v - row_left) * K;
    for (int k = 0; k < K; k++)
      nv_[k] = dv[k] + CF_GAMMA * (acc[k] - CF_LAMBDA * dv[k]);
  }
}

void cf_cpu(const HostCSC& g, int K, int iters, float* vec_out) {
  std::vector<float> oldv((uint64_t)g.nv * K), newv((uint64_t)g.nv * K);
  cf_init(g.nv, K, oldv.data());
  for (int it = 0; it < iters; it++) {
    cf_iter_part(0, g.nv - 1, 0, g.col_end.data(), g.src.data(),
                 g.weight.data(), K, oldv.data(), newv.data());
    std::swap(oldv, newv);
  }
  std::memcpy(vec_out, oldv.data(), sizeof(float) * (uint64_t)g.nv * K);
}

// CF training loss (for tests: loss must decrease over sweeps).
double cf_loss(const HostCSC& g, int K, const float* vec) {
  double loss = 0;
  for (V_ID v = 0; v < g.nv; v++) {
    const float* dv = vec + (uint64_t)v * K;
    for (E_ID i = g.row_begin(v); i < g.row_end(v); i++) {
      const float* sv = vec + (uint64_t)g.src[i] * K;
      float dot = 0.0f;
      for (int k = 0; k < K; k++) dot += sv[k] * dv[k];
      double err = (double)g.weight[i] - dot;
      loss += err * err;
    }
  }
  return loss;
}

}  // namespace lux
