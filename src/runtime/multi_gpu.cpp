// Native multi-GPU PageRank: one forked child per GPU, RCCL over xGMI.
//
// Design (matches the Python engine's exchange semantics, engine.py
// PagerankEngine, but PyTorch-free): each child owns device r, holds its
// edge-balanced CSC slice + the replicated rank vector, and per iteration
// runs the degree-binned pull sweep over its partition followed by an
// all-gather(v) of rank slices — grouped ncclSend/ncclRecv pairs, the
// direct point-to-point shape that fits the 7-link xGMI mesh (no ring).
// The reference's multi-GPU path instead bounced slices through zero-copy
// host memory under Legion (pagerank_gpu.cu:105-151).
#include <rccl/rccl.h>
#include <sys/wait.h>
#include <unistd.h>

#include <chrono>
#include <cstdio>
#include <cstring>
#include <string>
#include <unordered_map>
#include <vector>

#include "multi_gpu.h"

namespace lux {

#define LUX_NCCL(cmd)                                                       \
  do {                                                                      \
    ncclResult_t r_ = (cmd);                                                \
    if (r_ != ncclSuccess) {                                                \
      fprintf(stderr, "RCCL error %s:%d: %s\n", __FILE__, __LINE__,         \
              ncclGetErrorString(r_));                                      \
      _exit(13);                                                            \
    }                                                                       \
  } while (0)

namespace {

double now_seconds() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// LUXS vertex-state dump (lux_amd/checkpoint.py format)
void dump_state_device(const char* path, const void* dev_ptr, int dtype,
                       uint32_t k, uint32_t nv, uint64_t iter) {
  std::vector<char> host((size_t)nv * k * 4);
  LUX_OK(hipMemcpy(host.data(), dev_ptr, host.size(),
                   hipMemcpyDeviceToHost));
  FILE* f = fopen(path, "wb");
  if (!f) {
    perror(path);
    return;
  }
  uint32_t hdr[4] = {0x5358554Cu, (uint32_t)dtype, k, nv};
  fwrite(hdr, 4, 4, f);
  fwrite(&iter, 8, 1, f);
  fwrite(host.data(), 1, host.size(), f);
  fclose(f);
  printf("[lux] wrote %s (nv=%u k=%u)\n", path, nv, k);
}

}  // namespace

// Re-exec'd child entry (LUX_MULTI_RANK env): owns one device, joins the
// RCCL communicator via the id file rank 0 writes (ncclGetUniqueId
// initialises the HIP runtime, so it must never run in the launcher —
// forked children of a HIP-initialised parent segfault in HSA).
int pagerank_multi_child(const HostCSC& g, int rank, int ngpus,
                         const char* idfile, int iters, bool verbose,
                         const char* dump) {
  int ndev = 0;
  LUX_OK(hipGetDeviceCount(&ndev));
  if (rank >= ndev) {
    if (rank == ndev)  // one child reports
      fprintf(stderr, "[lux] -ll:gpu %d but only %d visible GPU(s)\n",
              ngpus, ndev);
    return 3;
  }
  LUX_OK(hipSetDevice(rank));
  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  ncclUniqueId id;
  if (rank == 0) {
    LUX_NCCL(ncclGetUniqueId(&id));
    std::string tmp = std::string(idfile) + ".tmp";
    FILE* f = fopen(tmp.c_str(), "wb");
    if (!f || fwrite(&id, sizeof(id), 1, f) != 1) {
      perror(idfile);
      return 4;
    }
    fclose(f);
    rename(tmp.c_str(), idfile);
  } else {
    FILE* f = nullptr;
    for (int tries = 0; tries < 1200 && !f; tries++) {  // <=120 s
      f = fopen(idfile, "rb");
      if (!f) usleep(100000);
    }
    if (!f || fread(&id, sizeof(id), 1, f) != 1) {
      fprintf(stderr, "[lux] rank %d: no RCCL id file %s\n", rank, idfile);
      return 4;
    }
    fclose(f);
  }
  ncclComm_t comm;
  LUX_NCCL(ncclCommInitRank(&comm, ngpus, id, rank));

  Partition part = partition_edge_balanced(g.nv, g.ne, g.col_end.data(),
                                           ngpus);
  V_ID rl = part.row_left[rank], rr = part.row_right[rank];
  V_ID vp = rr >= rl ? rr - rl + 1 : 0;
  E_ID cl = part.col_left[rank];
  E_ID ep = part.col_right[rank] - cl;
  std::vector<V_ID> verts(ngpus);
  for (int r = 0; r < ngpus; r++)
    verts[r] = part.row_right[r] >= part.row_left[r]
                   ? part.row_right[r] - part.row_left[r] + 1
                   : 0;

  size_t arena_bytes = 8ull * (vp + 1) + 8ull * vp + 4ull * ep  // slice
                       + 12ull * vp + (64ull << 20)             // bins
                       + 4ull * g.nv                            // degrees
                       + 4ull * g.nv + 4ull * vp                // old/new
                       + BlockedPull::arena_bytes(g.nv, vp, ep)
                       + (8ull << 20);
  DeviceArena arena(arena_bytes);
  // upload ONLY my slice (col_end slice + col slice); the replicated
  // rank vector is the one O(nv) array (the reference replicates it in
  // ZC host memory instead, core/pull_model.inl:454-461)
  E_ID* col_end_sl = arena.alloc_n<E_ID>(vp ? vp : 1);
  V_ID* col = arena.alloc_n<V_ID>(ep ? ep : 1);
  if (vp)
    LUX_OK(hipMemcpyAsync(col_end_sl, g.col_end.data() + rl,
                          sizeof(E_ID) * vp, hipMemcpyHostToDevice, s));
  if (ep)
    LUX_OK(hipMemcpyAsync(col, g.src.data() + cl, sizeof(V_ID) * ep,
                          hipMemcpyHostToDevice, s));
  E_ID* row_ptr = arena.alloc_n<E_ID>(vp + 1);
  lux_gpu_local_row_ptr((uint64_t)s, vp, cl, col_end_sl, row_ptr);
  Bins bins;
  bins.build(row_ptr, vp, ep, arena, s);
  BlockedPull blocked;
  blocked.build(row_ptr, col, vp, ep, g.nv, arena, s);
  // global out-degrees: slice histogram + RCCL all-reduce
  V_ID* deg = arena.alloc_n<V_ID>(g.nv);
  LUX_OK(hipMemsetAsync(deg, 0, sizeof(V_ID) * g.nv, s));
  lux_gpu_hist_u32((uint64_t)s, ep, col, deg);
  LUX_NCCL(ncclAllReduce(deg, deg, g.nv, ncclUint32, ncclSum, comm, s));
  float* old_ = arena.alloc_n<float>(g.nv);
  float* new_ = arena.alloc_n<float>(vp ? vp : 1);
  {
    std::vector<V_ID> hdeg(g.nv);
    LUX_OK(hipMemcpyAsync(hdeg.data(), deg, sizeof(V_ID) * g.nv,
                          hipMemcpyDeviceToHost, s));
    LUX_OK(hipStreamSynchronize(s));
    std::vector<float> hpr(g.nv);
    float r0 = 1.0f / g.nv;
    for (V_ID v = 0; v < g.nv; v++)
      hpr[v] = hdeg[v] == 0 ? r0 : r0 / hdeg[v];
    LUX_OK(hipMemcpyAsync(old_, hpr.data(), sizeof(float) * g.nv,
                          hipMemcpyHostToDevice, s));
    LUX_OK(hipStreamSynchronize(s));  // hpr leaves scope
  }
  float init_rank = (1.0f - PR_ALPHA) / g.nv;

  // barrier (tiny all-reduce) + sync brackets the timed loop
  float* bar = arena.alloc_n<float>(1);
  LUX_NCCL(ncclAllReduce(bar, bar, 1, ncclFloat, ncclSum, comm, s));
  LUX_OK(hipStreamSynchronize(s));
  double t0 = now_seconds();
  for (int it = 0; it < iters; it++) {
    if (vp) {
      LUX_OK(hipMemsetAsync(new_, 0, sizeof(float) * vp, s));
      if (blocked.active())
        blocked.sweep(0, old_, new_, deg, rl, init_rank, s);
      else
        lux_gpu_pull_iter((uint64_t)s, 0, bins.n0, bins.bin0, bins.n1,
                          bins.bin1, bins.n2, bins.bin2, bins.nbig,
                          bins.bin2v, row_ptr, 0, col, old_, new_, deg, rl,
                          init_rank);
      lux_gpu_pull_finish_pr((uint64_t)s, vp, new_, deg, rl, init_rank);
    }
    // all-gather(v) of slices: direct pairwise sends on the xGMI mesh
    LUX_NCCL(ncclGroupStart());
    for (int r = 0; r < ngpus; r++) {
      if (r == rank) continue;
      if (vp) LUX_NCCL(ncclSend(new_, vp, ncclFloat, r, comm, s));
      if (verts[r])
        LUX_NCCL(ncclRecv(old_ + part.row_left[r], verts[r], ncclFloat, r,
                          comm, s));
    }
    LUX_NCCL(ncclGroupEnd());
    if (vp)
      LUX_OK(hipMemcpyAsync(old_ + rl, new_, sizeof(float) * vp,
                            hipMemcpyDeviceToDevice, s));
  }
  LUX_NCCL(ncclAllReduce(bar, bar, 1, ncclFloat, ncclSum, comm, s));
  LUX_OK(hipStreamSynchronize(s));
  double secs = now_seconds() - t0;
  if (rank == 0) {
    printf("ELAPSED TIME = %7.7f s\n", secs);
    printf("[lux] %.3f GTEPS (%d iterations, %llu edges, %d GPUs)\n",
           double(g.ne) * iters / secs / 1e9, iters,
           (unsigned long long)g.ne, ngpus);
    if (dump) dump_state_device(dump, old_, 0, 1, g.nv, (uint64_t)iters);
    if (verbose) {
      float first[5];
      LUX_OK(hipMemcpy(first, old_, sizeof(first), hipMemcpyDeviceToHost));
      printf("[lux] first ranks (pr/out_degree): %g %g %g %g %g\n", first[0],
             first[1], first[2], first[3], first[4]);
    }
  }
  ncclCommDestroy(comm);
  return 0;
}

// ---- shared worker bootstrap (device + RCCL + slice upload) ----
namespace {

struct MultiCtx {
  hipStream_t s;
  ncclComm_t comm;
  Partition part;
  V_ID rl = 0, rr = 0, vp = 0;
  E_ID cl = 0, ep = 0;
  std::vector<V_ID> verts;
  E_ID* row_ptr = nullptr;  // local, 0-based
  V_ID* col = nullptr;
  WeightType* w = nullptr;
  Bins bins;
  BlockedPull blocked;
};

// device + communicator + partition (NO allocations — the caller builds
// its DeviceArena AFTER this call so it lands on the worker's device)
int multi_join(const HostCSC& g, int ngpus, int rank, const char* idfile,
               MultiCtx* c) {
  int ndev = 0;
  LUX_OK(hipGetDeviceCount(&ndev));
  if (rank >= ndev) {
    if (rank == ndev)
      fprintf(stderr, "[lux] -ll:gpu %d but only %d visible GPU(s)\n",
              ngpus, ndev);
    return 3;
  }
  LUX_OK(hipSetDevice(rank));
  LUX_OK(hipStreamCreate(&c->s));
  ncclUniqueId id;
  if (rank == 0) {
    LUX_NCCL(ncclGetUniqueId(&id));
    std::string tmp = std::string(idfile) + ".tmp";
    FILE* f = fopen(tmp.c_str(), "wb");
    if (!f || fwrite(&id, sizeof(id), 1, f) != 1) {
      perror(idfile);
      return 4;
    }
    fclose(f);
    rename(tmp.c_str(), idfile);
  } else {
    FILE* f = nullptr;
    for (int tries = 0; tries < 1200 && !f; tries++) {
      f = fopen(idfile, "rb");
      if (!f) usleep(100000);
    }
    if (!f || fread(&id, sizeof(id), 1, f) != 1) {
      fprintf(stderr, "[lux] rank %d: no RCCL id file %s\n", rank, idfile);
      return 4;
    }
    fclose(f);
  }
  LUX_NCCL(ncclCommInitRank(&c->comm, ngpus, id, rank));
  c->part = partition_edge_balanced(g.nv, g.ne, g.col_end.data(), ngpus);
  c->rl = c->part.row_left[rank];
  c->rr = c->part.row_right[rank];
  c->vp = c->rr >= c->rl ? c->rr - c->rl + 1 : 0;
  c->cl = c->part.col_left[rank];
  c->ep = c->part.col_right[rank] - c->cl;
  c->verts.resize(ngpus);
  for (int r = 0; r < ngpus; r++)
    c->verts[r] = c->part.row_right[r] >= c->part.row_left[r]
                      ? c->part.row_right[r] - c->part.row_left[r] + 1
                      : 0;
  return 0;
}

// upload ONLY my slice (call after the arena exists on my device).
// want_blocked: build the src-blocked CSC (only the push worker's pull
// fallback sweeps use it; CC/CF workers skip the memory)
void multi_upload(const HostCSC& g, bool weighted, DeviceArena& arena,
                  MultiCtx* c, bool want_blocked = false) {
  E_ID* col_end_sl = arena.alloc_n<E_ID>(c->vp ? c->vp : 1);
  c->col = arena.alloc_n<V_ID>(c->ep ? c->ep : 1);
  if (c->vp)
    LUX_OK(hipMemcpyAsync(col_end_sl, g.col_end.data() + c->rl,
                          sizeof(E_ID) * c->vp, hipMemcpyHostToDevice,
                          c->s));
  if (c->ep)
    LUX_OK(hipMemcpyAsync(c->col, g.src.data() + c->cl,
                          sizeof(V_ID) * c->ep, hipMemcpyHostToDevice,
                          c->s));
  if (weighted && c->ep) {
    c->w = arena.alloc_n<WeightType>(c->ep);
    LUX_OK(hipMemcpyAsync(c->w, g.weight.data() + c->cl,
                          sizeof(WeightType) * c->ep, hipMemcpyHostToDevice,
                          c->s));
  }
  c->row_ptr = arena.alloc_n<E_ID>(c->vp + 1);
  lux_gpu_local_row_ptr((uint64_t)c->s, c->vp, c->cl, col_end_sl,
                        c->row_ptr);
  c->bins.build(c->row_ptr, c->vp, c->ep, arena, c->s);
  if (want_blocked)
    c->blocked.build(c->row_ptr, c->col, c->vp, c->ep, g.nv, arena, c->s);
}

void multi_barrier(MultiCtx& c, float* bar) {
  LUX_NCCL(ncclAllReduce(bar, bar, 1, ncclFloat, ncclSum, c.comm, c.s));
  LUX_OK(hipStreamSynchronize(c.s));
}

}  // namespace

int components_multi_child(const HostCSC& g, int rank, int ngpus,
                           const char* idfile, bool check, const char* dump,
                           bool verbose) {
  MultiCtx c;
  int rc = multi_join(g, ngpus, rank, idfile, &c);
  if (rc) return rc;
  DeviceArena arena(16ull * g.nv                     // parent + labels
                    + (uint64_t)ngpus * 4ull * g.nv  // gathered stars
                    + 8ull * g.nv                    // col_end slice
                    + 8ull * (g.ne / (ngpus ? ngpus : 1) + 1)
                    + 12ull * g.nv + (96ull << 20));
  multi_upload(g, false, arena, &c);
  V_ID* parent = arena.alloc_n<V_ID>(g.nv);
  V_ID* labels = arena.alloc_n<V_ID>(g.nv);
  V_ID* gathered = arena.alloc_n<V_ID>((uint64_t)ngpus * g.nv);
  uint32_t* gbits = arena.alloc_n<uint32_t>((g.nv + 31) / 32);
  unsigned long long* diff = arena.alloc_n<unsigned long long>(1);
  float* bar = arena.alloc_n<float>(1);
  {
    std::vector<V_ID> h(g.nv);
    for (V_ID v = 0; v < g.nv; v++) h[v] = v;
    LUX_OK(hipMemcpyAsync(parent, h.data(), 4ull * g.nv,
                          hipMemcpyHostToDevice, c.s));
    LUX_OK(hipStreamSynchronize(c.s));  // h leaves scope
  }
  multi_barrier(c, bar);
  double t0 = now_seconds();
  // Afforest sample-hook + giant-skip sweep over MY edge slice
  // (cc_engine.py run, per rank)
  for (uint32_t k = 0; k < 2; k++)
    lux_gpu_uf_union_kth((uint64_t)c.s, c.vp, c.row_ptr, c.col, c.rl,
                         parent, k);
  lux_gpu_uf_flatten((uint64_t)c.s, g.nv, parent, labels);
  V_ID giant;
  {
    std::vector<V_ID> hl(g.nv);
    LUX_OK(hipMemcpyAsync(hl.data(), labels, 4ull * g.nv,
                          hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    V_ID stride = g.nv / 65536 ? g.nv / 65536 : 1;
    std::unordered_map<V_ID, uint32_t> freq;
    V_ID best = hl[0];
    uint32_t bestc = 0;
    for (V_ID v = 0; v < g.nv; v += stride) {
      uint32_t n = ++freq[hl[v]];
      if (n > bestc) {
        bestc = n;
        best = hl[v];
      }
    }
    giant = best;
  }
  lux_gpu_cc_giant_bits((uint64_t)c.s, g.nv, labels, giant, gbits);
  lux_gpu_uf_union_binned((uint64_t)c.s, c.bins.n0, c.bins.bin0, c.bins.n1,
                          c.bins.bin1, c.bins.n2, c.bins.bin2, c.row_ptr,
                          c.col, c.rl, parent, gbits);
  lux_gpu_uf_flatten((uint64_t)c.s, g.nv, parent, labels);
  // star-forest exchange: allgather full label vectors, union peers'
  // stars, repeat until no label moves (monotone merges -> O(log P))
  int rounds = 1;
  while (ngpus > 1) {
    LUX_NCCL(ncclAllGather(labels, gathered, g.nv, ncclUint32, c.comm,
                           c.s));
    for (int q = 0; q < ngpus; q++)
      if (q != rank)
        lux_gpu_uf_union_star((uint64_t)c.s, g.nv,
                              gathered + (uint64_t)q * g.nv, parent);
    lux_gpu_uf_flatten((uint64_t)c.s, g.nv, parent, labels);
    LUX_OK(hipMemsetAsync(diff, 0, 8, c.s));
    // prev labels live in my gathered copy
    lux_gpu_count_diff((uint64_t)c.s, g.nv,
                       gathered + (uint64_t)rank * g.nv, labels, diff);
    LUX_NCCL(ncclAllReduce(diff, diff, 1, ncclUint64, ncclSum, c.comm,
                           c.s));
    unsigned long long hd;
    LUX_OK(hipMemcpyAsync(&hd, diff, 8, hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    rounds++;
    if (hd == 0) break;
  }
  multi_barrier(c, bar);
  double secs = now_seconds() - t0;
  uint64_t mistakes = 0;
  if (check) {
    unsigned long long* m = arena.alloc_n<unsigned long long>(1);
    LUX_OK(hipMemsetAsync(m, 0, 8, c.s));
    lux_gpu_check((uint64_t)c.s, 0, c.vp, c.rl, c.row_ptr, c.col, labels,
                  m);
    LUX_NCCL(ncclAllReduce(m, m, 1, ncclUint64, ncclSum, c.comm, c.s));
    unsigned long long hm;
    LUX_OK(hipMemcpyAsync(&hm, m, 8, hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    mistakes = hm;
  }
  if (rank == 0) {
    printf("ELAPSED TIME = %7.7f s\n", secs);
    if (dump) dump_state_device(dump, labels, 1, 1, g.nv, (uint64_t)rounds);
    printf("[lux] converged in %d exchange rounds, %.3f GTEPS (%d GPUs)\n",
           rounds, double(g.ne) / secs / 1e9, ngpus);
    if (check)
      printf("[%s] %llu mistakes\n", mistakes == 0 ? "PASS" : "FAIL",
             (unsigned long long)mistakes);
  }
  (void)verbose;
  ncclCommDestroy(c.comm);
  return check && mistakes ? 1 : 0;
}

int col_filter_multi_child(const HostCSC& g, int rank, int ngpus,
                           const char* idfile, int K, int iters, bool als,
                           const char* dump, V_ID n_users) {
  MultiCtx c;
  int rc = multi_join(g, ngpus, rank, idfile, &c);
  if (rc) return rc;
  DeviceArena arena(
      8ull * g.nv + 8ull * (g.ne / (ngpus ? ngpus : 1) + 1) * 2
      + 8ull * (uint64_t)g.nv * K  // old + new slack
      + 12ull * g.nv + (96ull << 20)
      + (als ? 4ull * g.nv + (17ull << 20) * (g.ne / 2048 / 1024 + 1) : 0));
  multi_upload(g, true, arena, &c);
  float* old_ = arena.alloc_n<float>((uint64_t)g.nv * K);
  float* new_ = arena.alloc_n<float>((uint64_t)(c.vp ? c.vp : 1) * K);
  {
    std::vector<float> h((uint64_t)g.nv * K, sqrtf(1.0f / K));
    if (als)  // jittered init (see als_init_val); SGD keeps the constant
      for (size_t i = 0; i < h.size(); i++) h[i] = als_init_val(i, K);
    LUX_OK(hipMemcpyAsync(old_, h.data(), 4ull * g.nv * K,
                          hipMemcpyHostToDevice, c.s));
    LUX_OK(hipStreamSynchronize(c.s));  // h leaves scope
  }
  int* hubidx = nullptr;
  float *gram = nullptr, *rhs = nullptr;
  // Gauss-Seidel alternation (n_users > 0): split my bin lists at the
  // LOCAL user/item boundary; hub scratch slots become phase-relative
  // (see SingleGpuCF). A rank whose partition is all-items still joins
  // both publish collectives with an empty user half-sweep.
  bool alt = als && n_users > 0;
  BinSplit split;
  if (alt && c.vp) {
    V_ID lb = n_users <= c.rl ? 0
              : (n_users - c.rl < c.vp ? n_users - c.rl : c.vp);
    split = split_bins_at(c.bins, lb, c.s);
  }
  if (als && c.bins.nbig) {
    hubidx = arena.alloc_n<int>(c.vp ? c.vp : 1);
    gram = arena.alloc_n<float>((uint64_t)c.bins.nbig * 64 * 64);
    rhs = arena.alloc_n<float>((uint64_t)c.bins.nbig * 64);
    std::vector<V_ID> hv(c.bins.nbig);
    LUX_OK(hipMemcpyAsync(hv.data(), c.bins.bin2v, 4ull * c.bins.nbig,
                          hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    std::vector<int> hidx(c.vp, -1);
    for (uint32_t i = 0; i < c.bins.nbig; i++)
      hidx[hv[i]] = (int)(alt && i >= split.nbigu ? i - split.nbigu : i);
    LUX_OK(hipMemcpyAsync(hubidx, hidx.data(), 4ull * c.vp,
                          hipMemcpyHostToDevice, c.s));
    LUX_OK(hipStreamSynchronize(c.s));  // hidx leaves scope
  }
  float* bar = arena.alloc_n<float>(1);
  multi_barrier(c, bar);
  double t0 = now_seconds();
  // all-gather(v) of K-dim vector slices (direct pairwise over xGMI)
  auto publish = [&]() {
    LUX_NCCL(ncclGroupStart());
    for (int r = 0; r < ngpus; r++) {
      if (r == rank) continue;
      if (c.vp)
        LUX_NCCL(ncclSend(new_, (uint64_t)c.vp * K, ncclFloat, r, c.comm,
                          c.s));
      if (c.verts[r])
        LUX_NCCL(ncclRecv(old_ + (uint64_t)c.part.row_left[r] * K,
                          (uint64_t)c.verts[r] * K, ncclFloat, r, c.comm,
                          c.s));
    }
    LUX_NCCL(ncclGroupEnd());
    if (c.vp)
      LUX_OK(hipMemcpyAsync(old_ + (uint64_t)c.rl * K, new_,
                            4ull * c.vp * K, hipMemcpyDeviceToDevice, c.s));
  };
  for (int it = 0; it < iters; it++) {
    if (c.vp && als) {
      LUX_OK(hipMemcpyAsync(new_, old_ + (uint64_t)c.rl * K,
                            4ull * c.vp * K, hipMemcpyDeviceToDevice,
                            c.s));
      if (c.bins.nbig) {
        LUX_OK(hipMemsetAsync(gram, 0, 4ull * c.bins.nbig * 64 * 64,
                              c.s));
        LUX_OK(hipMemsetAsync(rhs, 0, 4ull * c.bins.nbig * 64, c.s));
      }
    }
    if (alt) {
      // user half-sweep against old item factors, publish, then item
      // half-sweep against the globally UPDATED users in old_
      if (c.vp)
        lux_gpu_cf_als_iter((uint64_t)c.s, split.n0u, c.bins.bin0,
                            split.n1u, c.bins.bin1, split.n2u, c.bins.bin2,
                            split.nbigu, c.bins.bin2v, hubidx, gram, rhs,
                            c.row_ptr, c.col, c.w, old_, nullptr, new_,
                            c.rl, K);
      publish();
      if (c.vp)
        lux_gpu_cf_als_iter(
            (uint64_t)c.s, c.bins.n0 - split.n0u, c.bins.bin0 + split.n0u,
            c.bins.n1 - split.n1u, c.bins.bin1 + split.n1u,
            c.bins.n2 - split.n2u, c.bins.bin2 + split.n2u,
            c.bins.nbig - split.nbigu, c.bins.bin2v + split.nbigu, hubidx,
            gram ? gram + (uint64_t)split.nbigu * 64 * 64 : nullptr,
            rhs ? rhs + (uint64_t)split.nbigu * 64 : nullptr, c.row_ptr,
            c.col, c.w, old_, nullptr, new_, c.rl, K);
      publish();
      continue;
    }
    if (c.vp) {
      if (als) {
        lux_gpu_cf_als_iter((uint64_t)c.s, c.bins.n0, c.bins.bin0,
                            c.bins.n1, c.bins.bin1, c.bins.n2, c.bins.bin2,
                            c.bins.nbig, c.bins.bin2v, hubidx, gram, rhs,
                            c.row_ptr, c.col, c.w, old_, nullptr, new_,
                            c.rl, K);
      } else {
        lux_gpu_cf_seed((uint64_t)c.s, (uint64_t)c.vp * K,
                        old_ + (uint64_t)c.rl * K, new_);
        lux_gpu_cf_iter((uint64_t)c.s, c.bins.n0, c.bins.bin0, c.bins.n1,
                        c.bins.bin1, c.bins.n2, c.bins.bin2, c.bins.nbig,
                        c.bins.bin2v, c.row_ptr, c.col, c.w, old_, new_,
                        c.rl, K);
      }
    }
    publish();
  }
  multi_barrier(c, bar);
  double secs = now_seconds() - t0;
  if (rank == 0) {
    printf("ELAPSED TIME = %7.7f s\n", secs);
    if (dump)
      dump_state_device(dump, old_, 0, (uint32_t)K, g.nv, (uint64_t)iters);
    printf("[lux] %.3f GTEPS (%d sweeps, rank %d, %d GPUs)\n",
           double(g.ne) * iters / secs / 1e9, iters, K, ngpus);
  }
  ncclCommDestroy(c.comm);
  return 0;
}

// ---- native multi-GPU push engine (SSSP hop / CC label prop) ----
// The C++/RCCL twin of lux_amd/push_engine.py: per iteration exactly ONE
// blocking host read (the all-gathered 32 B/rank meta record); dense/
// sparse conversions run in the device fixup chain; payloads ship only
// USED bytes; label slices travel only on dense iterations (sparse
// queues carry the label annex, repaired into the replicated array by
// frontier_expand).
int push_multi_child(const HostCSC& g, int rank, int ngpus, bool is_min,
                     V_ID source, const char* idfile, bool check,
                     const char* dump, bool verbose) {
  MultiCtx c;
  int rc = multi_join(g, ngpus, rank, idfile, &c);
  if (rc) return rc;
  // seg/annex layout for all ranks
  std::vector<uint64_t> seg_bytes(ngpus), seg_off(ngpus + 1, 0);
  std::vector<uint64_t> annex_cap(ngpus), annex_off(ngpus + 1, 0);
  for (int q = 0; q < ngpus; q++) {
    seg_bytes[q] = (frontier_bytes(c.verts[q]) + 15) & ~15ull;
    seg_off[q + 1] = seg_off[q] + seg_bytes[q];
    annex_cap[q] = frontier_capacity(c.verts[q]);
    annex_off[q + 1] = annex_off[q] + annex_cap[q];
  }
  V_ID capacity = frontier_capacity(c.vp);
  uint64_t max_items = (uint64_t)g.nv / SPARSE_THRESHOLD +
                       (c.ep ? c.ep : 1) / 8192 + 1024;
  DeviceArena arena(
      4ull * g.nv * 3                  // labels + deg + slack
      + 8ull * (g.nv + 2)              // push row_ptr
      + 4ull * (c.ep + 1) * 3          // pull col + push col + cursor slack
      + 8ull * (g.nv + 2)              // col_end slice + cursor
      + 12ull * g.nv + (96ull << 20)   // bins + slack
      + seg_off[ngpus] + 3 * seg_bytes[rank] + 4 * annex_off[ngpus]
      + 16ull * max_items
      + BlockedPull::arena_bytes(g.nv, c.vp, c.ep) + (16ull << 20));
  multi_upload(g, false, arena, &c, /*want_blocked=*/true);
  // push CSR: all nv sources -> my-partition dsts
  E_ID* push_row_ptr = arena.alloc_n<E_ID>(g.nv + 1);
  V_ID* push_col = arena.alloc_n<V_ID>(c.ep ? c.ep : 1);
  V_ID* deg = arena.alloc_n<V_ID>(g.nv);  // global out-degrees after
                                          // allreduce; slice at +rl
  {
    E_ID* ends;
    unsigned long long *cursor, *partials;
    LUX_OK(hipMalloc(&ends, sizeof(E_ID) * g.nv));
    LUX_OK(hipMalloc(&cursor, sizeof(uint64_t) * g.nv));
    LUX_OK(hipMalloc(&partials,
                     sizeof(uint64_t) * lux_gpu_scan_partials_size(g.nv)));
    LUX_OK(hipMemsetAsync(deg, 0, sizeof(V_ID) * g.nv, c.s));
    lux_gpu_hist_u32((uint64_t)c.s, c.ep, c.col, deg);
    lux_gpu_scan_end_offsets((uint64_t)c.s, g.nv, deg, ends, partials);
    lux_gpu_local_row_ptr((uint64_t)c.s, g.nv, 0, ends, push_row_ptr);
    LUX_OK(hipMemcpyAsync(cursor, push_row_ptr, sizeof(E_ID) * g.nv,
                          hipMemcpyDeviceToDevice, c.s));
    lux_gpu_csr_scatter((uint64_t)c.s, c.ep, c.col, c.row_ptr, c.vp, c.rl,
                        cursor, push_col);
    // global out-degrees (the meta edge-volume pricing input)
    LUX_NCCL(ncclAllReduce(deg, deg, g.nv, ncclUint32, ncclSum, c.comm,
                           c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    hipFree(ends);
    hipFree(cursor);
    hipFree(partials);
  }
  uint32_t* labels = arena.alloc_n<uint32_t>(g.nv);
  uint32_t* labels_part = arena.alloc_n<uint32_t>(c.vp ? c.vp : 1);
  uint32_t* snapshot = arena.alloc_n<uint32_t>(c.vp ? c.vp : 1);
  uint8_t* fq_all = arena.alloc_n<uint8_t>(seg_off[ngpus]);
  uint8_t* new_seg = arena.alloc_n<uint8_t>(seg_bytes[rank]);
  uint8_t* tmp_seg = arena.alloc_n<uint8_t>(seg_bytes[rank]);
  uint32_t* annex_all = arena.alloc_n<uint32_t>(annex_off[ngpus]);
  uint32_t* new_annex = arena.alloc_n<uint32_t>(capacity ? capacity : 1);
  lux_uint2* items = arena.alloc_n<lux_uint2>(max_items);
  uint32_t* item_counter = arena.alloc_n<uint32_t>(4);
  uint32_t* meta_mine = arena.alloc_n<uint32_t>(8);
  uint32_t* meta_all = arena.alloc_n<uint32_t>(8ull * ngpus);
  uint32_t* bits = nullptr;
  if (is_min) bits = arena.alloc_n<uint32_t>((c.vp + 31) / 32 + 1);
  bool bits_stale = true;
  float* bar = arena.alloc_n<float>(1);

  // seed labels + frontier + meta (sssp_gpu.cu:733-744 / components)
  std::vector<uint32_t> meta_h(8ull * ngpus, 0);
  {
    std::vector<uint32_t> hl(g.nv);
    std::vector<uint8_t> hfq(seg_off[ngpus], 0);
    for (int q = 0; q < ngpus; q++) {
      FrontierHeader* h = (FrontierHeader*)(hfq.data() + seg_off[q]);
      if (is_min) {
        h->type = FrontierHeader::SPARSE_QUEUE;
        bool owner = c.verts[q] && c.part.row_left[q] <= source &&
                     source <= c.part.row_right[q];
        h->numNodes = owner ? 1 : 0;
        if (owner)
          *(V_ID*)(hfq.data() + seg_off[q] + sizeof(FrontierHeader)) =
              source;
      } else {
        h->type = FrontierHeader::DENSE_BITMAP;
        h->numNodes = c.verts[q];
        memset(hfq.data() + seg_off[q] + sizeof(FrontierHeader), 0xFF,
               (c.verts[q] + 7) / 8);
      }
      meta_h[8ull * q + 0] = h->type;
      meta_h[8ull * q + 1] = h->numNodes;
    }
    if (is_min) {
      for (V_ID v = 0; v < g.nv; v++) hl[v] = INF_LABEL;
      hl[source] = 0;
    } else {
      for (V_ID v = 0; v < g.nv; v++) hl[v] = v;
    }
    LUX_OK(hipMemcpyAsync(labels, hl.data(), 4ull * g.nv,
                          hipMemcpyHostToDevice, c.s));
    if (c.vp)
      LUX_OK(hipMemcpyAsync(labels_part, hl.data() + c.rl, 4ull * c.vp,
                            hipMemcpyHostToDevice, c.s));
    LUX_OK(hipMemcpyAsync(fq_all, hfq.data(), seg_off[ngpus],
                          hipMemcpyHostToDevice, c.s));
    LUX_OK(hipMemsetAsync(annex_all, 0, 4ull * annex_off[ngpus], c.s));
    // drain before hl/hfq leave scope (async copy from pageable host)
    LUX_OK(hipStreamSynchronize(c.s));
  }
  bool labels_current = true;
  auto sync_labels = [&]() {
    if (labels_current) return;
    LUX_NCCL(ncclGroupStart());
    for (int r = 0; r < ngpus; r++) {
      if (r == rank) continue;
      if (c.vp)
        LUX_NCCL(ncclSend(labels_part, c.vp, ncclUint32, r, c.comm, c.s));
      if (c.verts[r])
        LUX_NCCL(ncclRecv(labels + c.part.row_left[r], c.verts[r],
                          ncclUint32, r, c.comm, c.s));
    }
    LUX_NCCL(ncclGroupEnd());
    if (c.vp)
      LUX_OK(hipMemcpyAsync(labels + c.rl, labels_part, 4ull * c.vp,
                            hipMemcpyDeviceToDevice, c.s));
    labels_current = true;
  };

  multi_barrier(c, bar);
  double t0 = now_seconds();
  int iters = 0;
  while (true) {
    // decisions from the CURRENT frontier's meta
    uint64_t old_fq = 0, evol = 0;
    int dense_votes = 0;
    bool overflow = false;
    for (int q = 0; q < ngpus; q++) {
      old_fq += meta_h[8ull * q + 1];
      evol += ((uint64_t)meta_h[8ull * q + 3] << 32) | meta_h[8ull * q + 2];
      if (meta_h[8ull * q + 0] == FrontierHeader::DENSE_BITMAP)
        dense_votes++;
      if (meta_h[8ull * q + 4]) overflow = true;
    }
    bool new_dense = dense_votes >= ngpus - dense_votes;
    if (c.vp) {
      LUX_OK(hipMemcpyAsync(snapshot, labels_part, 4ull * c.vp,
                            hipMemcpyDeviceToDevice, c.s));
      LUX_OK(hipMemsetAsync(new_seg, 0, sizeof(FrontierHeader), c.s));
    }
    bool pull_fallback =
        overflow || old_fq > (uint64_t)g.nv / SPARSE_THRESHOLD;
    if (overflow && rank == 0)
      fprintf(stderr, "[lux] frontier expand overflow: forced pull\n");
    bool did_push = false;
    if (!pull_fallback) {
      uint64_t thresh = is_min ? g.ne / 2 : g.ne / 8;
      if (evol > thresh) pull_fallback = true;
      else if ((evol * (c.ep ? c.ep : 1) / g.ne) / 16 > capacity)
        new_dense = true;
    }
    if (pull_fallback) {
      new_dense = true;
      sync_labels();
      if (c.vp) {
        LUX_OK(hipMemcpyAsync(labels_part, labels + c.rl, 4ull * c.vp,
                              hipMemcpyDeviceToDevice, c.s));
        if (c.blocked.active())
          c.blocked.sweep(is_min ? 1 : 2, labels, labels_part, nullptr,
                          c.rl, 0.0f, c.s);
        else
          lux_gpu_pull_iter((uint64_t)c.s, is_min ? 1 : 2, c.bins.n0,
                            c.bins.bin0, c.bins.n1, c.bins.bin1, c.bins.n2,
                            c.bins.bin2, c.bins.nbig, c.bins.bin2v,
                            c.row_ptr, 0, c.col, labels, labels_part,
                            nullptr, c.rl, 0.0f);
      }
      bits_stale = true;
    } else {
      did_push = true;
      LUX_OK(hipMemsetAsync(item_counter, 0, 16, c.s));
      for (int q = 0; q < ngpus; q++) {
        uint32_t typ = meta_h[8ull * q + 0], num = meta_h[8ull * q + 1];
        if (!c.verts[q]) continue;
        if (typ == FrontierHeader::DENSE_BITMAP) {
          lux_gpu_frontier_expand((uint64_t)c.s, 1, c.part.row_left[q],
                                  c.verts[q], fq_all + seg_off[q], nullptr,
                                  nullptr, push_row_ptr, items,
                                  item_counter, (uint32_t)max_items);
        } else if (num) {
          lux_gpu_frontier_expand((uint64_t)c.s, 0, 0, num,
                                  fq_all + seg_off[q],
                                  annex_all + annex_off[q], labels,
                                  push_row_ptr, items, item_counter,
                                  (uint32_t)max_items);
        }
      }
      uint32_t* b = nullptr;
      if (is_min && c.vp) {
        if (bits_stale) {
          lux_gpu_bits_from_labels((uint64_t)c.s, c.vp, labels_part, bits);
          bits_stale = false;
        }
        b = bits;
      }
      lux_gpu_push_chunk_scatter((uint64_t)c.s, is_min ? 1 : 0,
                                 new_dense ? 1 : 0, items, item_counter,
                                 (uint32_t)max_items, push_row_ptr,
                                 push_col, labels, snapshot, labels_part,
                                 c.rl, new_seg, capacity, b);
    }
    if (c.vp && new_dense)
      lux_gpu_build_bitmap((uint64_t)c.s, c.vp, snapshot, labels_part,
                           new_seg);
    lux_gpu_frontier_fixup((uint64_t)c.s, c.vp, c.rl, capacity,
                           new_dense ? 1 : 0, snapshot, labels_part,
                           deg + c.rl, new_seg, new_annex, tmp_seg,
                           meta_mine, did_push ? item_counter : nullptr,
                           (uint32_t)max_items);
    // meta first (the ONE host read), then payloads sized from it
    LUX_NCCL(ncclAllGather(meta_mine, meta_all, 8, ncclUint32, c.comm,
                           c.s));
    LUX_OK(hipMemcpyAsync(meta_h.data(), meta_all, 32ull * ngpus,
                          hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    // payload exchange: used bytes + annex + (dense ranks') label slices
    uint64_t used_me = 0, annex_me = 0;
    bool dense_me = meta_h[8ull * rank + 0] == FrontierHeader::DENSE_BITMAP;
    if (c.vp) {
      if (dense_me) used_me = 8 + ((uint64_t)c.vp + 7) / 8;
      else {
        used_me = 8 + 4ull * meta_h[8ull * rank + 1];
        annex_me = meta_h[8ull * rank + 1];
      }
    }
    LUX_NCCL(ncclGroupStart());
    for (int r = 0; r < ngpus; r++) {
      if (r == rank) continue;
      if (used_me)
        LUX_NCCL(ncclSend(new_seg, used_me, ncclUint8, r, c.comm, c.s));
      if (annex_me)
        LUX_NCCL(ncclSend(new_annex, annex_me, ncclUint32, r, c.comm,
                          c.s));
      if (dense_me && c.vp)
        LUX_NCCL(ncclSend(labels_part, c.vp, ncclUint32, r, c.comm, c.s));
      if (c.verts[r]) {
        bool dense_r =
            meta_h[8ull * r + 0] == FrontierHeader::DENSE_BITMAP;
        uint64_t used_r = dense_r ? 8 + ((uint64_t)c.verts[r] + 7) / 8
                                  : 8 + 4ull * meta_h[8ull * r + 1];
        LUX_NCCL(ncclRecv(fq_all + seg_off[r], used_r, ncclUint8, r,
                          c.comm, c.s));
        if (!dense_r && meta_h[8ull * r + 1])
          LUX_NCCL(ncclRecv(annex_all + annex_off[r],
                            meta_h[8ull * r + 1], ncclUint32, r, c.comm,
                            c.s));
        if (dense_r)
          LUX_NCCL(ncclRecv(labels + c.part.row_left[r], c.verts[r],
                            ncclUint32, r, c.comm, c.s));
      }
    }
    LUX_NCCL(ncclGroupEnd());
    // my own copies
    if (c.vp) {
      LUX_OK(hipMemcpyAsync(fq_all + seg_off[rank], new_seg, used_me,
                            hipMemcpyDeviceToDevice, c.s));
      if (annex_me)
        LUX_OK(hipMemcpyAsync(annex_all + annex_off[rank], new_annex,
                              4ull * annex_me, hipMemcpyDeviceToDevice,
                              c.s));
      if (dense_me)
        LUX_OK(hipMemcpyAsync(labels + c.rl, labels_part, 4ull * c.vp,
                              hipMemcpyDeviceToDevice, c.s));
    }
    // slice q fresh iff published now, or fresh before and unchanged
    bool all_ok = true;
    for (int q = 0; q < ngpus; q++) {
      if (!c.verts[q]) continue;
      bool pub = meta_h[8ull * q + 0] == FrontierHeader::DENSE_BITMAP;
      if (labels_current) {
        if (!pub && meta_h[8ull * q + 1]) all_ok = false;
      } else if (!pub) {
        all_ok = false;
      }
    }
    labels_current = all_ok;
    iters++;
    uint64_t total = 0;
    bool ovf = false;
    for (int q = 0; q < ngpus; q++) {
      total += meta_h[8ull * q + 1];
      if (meta_h[8ull * q + 4]) ovf = true;
    }
    if (verbose && rank == 0)
      printf("iter %d: activeNodes(%llu)\n", iters,
             (unsigned long long)total);
    if (total == 0 && !ovf) break;
    if ((uint64_t)iters > 4ull * g.nv) break;  // safety
  }
  sync_labels();
  multi_barrier(c, bar);
  double secs = now_seconds() - t0;
  uint64_t mistakes = 0;
  if (check) {
    unsigned long long* m = arena.alloc_n<unsigned long long>(1);
    LUX_OK(hipMemsetAsync(m, 0, 8, c.s));
    lux_gpu_check((uint64_t)c.s, is_min ? 1 : 0, c.vp, c.rl, c.row_ptr,
                  c.col, labels, m);
    LUX_NCCL(ncclAllReduce(m, m, 1, ncclUint64, ncclSum, c.comm, c.s));
    unsigned long long hm;
    LUX_OK(hipMemcpyAsync(&hm, m, 8, hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    mistakes = hm;
  }
  if (rank == 0) {
    printf("ELAPSED TIME = %7.7f s\n", secs);
    if (dump) dump_state_device(dump, labels, 1, 1, g.nv, (uint64_t)iters);
    printf("[lux] converged in %d iterations, %.3f GTEPS (%d GPUs)\n",
           iters, double(g.ne) / secs / 1e9, ngpus);
    if (check)
      printf("[%s] %llu mistakes\n", mistakes == 0 ? "PASS" : "FAIL",
             (unsigned long long)mistakes);
  }
  ncclCommDestroy(c.comm);
  return check && mistakes ? 1 : 0;
}

// Launcher: fork + EXEC one child per GPU (a plain fork would inherit the
// parent's process state; exec gives each rank a fresh runtime). The
// parent itself never touches HIP or RCCL.
int run_multi_workers(int ngpus, int argc, char** argv) {
  char idfile[64];
  snprintf(idfile, sizeof(idfile), "/tmp/lux_rccl_%d.id", (int)getpid());
  unlink(idfile);
  char exe[4096] = {0};
  ssize_t n = readlink("/proc/self/exe", exe, sizeof(exe) - 1);
  if (n <= 0) {
    perror("readlink");
    return 1;
  }
  std::vector<pid_t> pids;
  for (int r = 0; r < ngpus; r++) {
    pid_t p = fork();
    if (p < 0) {
      perror("fork");
      return 1;
    }
    if (p == 0) {
      char rs[16], ws[16];
      snprintf(rs, sizeof(rs), "%d", r);
      snprintf(ws, sizeof(ws), "%d", ngpus);
      setenv("LUX_MULTI_RANK", rs, 1);
      setenv("LUX_MULTI_WORLD", ws, 1);
      setenv("LUX_MULTI_IDFILE", idfile, 1);
      std::vector<char*> cargs;
      for (int i = 0; i < argc; i++) cargs.push_back(argv[i]);
      cargs.push_back(nullptr);
      execv(exe, cargs.data());
      perror("execv");
      _exit(127);
    }
    pids.push_back(p);
  }
  int rc = 0;
  for (pid_t p : pids) {
    int st = 0;
    waitpid(p, &st, 0);
    if (WIFEXITED(st) && WEXITSTATUS(st)) rc = WEXITSTATUS(st);
    if (WIFSIGNALED(st)) rc = 128 + WTERMSIG(st);
  }
  unlink(idfile);
  return rc;
}

int run_pagerank_multi(int ngpus, int argc, char** argv) {
  return run_multi_workers(ngpus, argc, argv);
}

int exec_torchrun_app(const char* module, int ngpus, int argc, char** argv) {
  // repo root = dirname(dirname(/proc/self/exe)) — binaries live in bin/
  char exe[4096] = {0};
  ssize_t n = readlink("/proc/self/exe", exe, sizeof(exe) - 1);
  std::string root = ".";
  if (n > 0) {
    std::string p(exe, n);
    size_t a = p.rfind('/');
    if (a != std::string::npos) {
      size_t b = p.rfind('/', a - 1);
      if (b != std::string::npos) root = p.substr(0, b);
    }
  }
  const char* old_pp = getenv("PYTHONPATH");
  std::string pp = old_pp ? root + ":" + old_pp : root;
  setenv("PYTHONPATH", pp.c_str(), 1);
  std::vector<std::string> args = {
      "python3", "-m", "torch.distributed.run", "--nnodes=1",
      "--nproc-per-node=" + std::to_string(ngpus),
      "--master-addr=127.0.0.1",
      "--master-port=" + std::to_string(29400 + (int)(getpid() % 1000)),
      "-m", module};
  for (int i = 1; i < argc; i++) args.push_back(argv[i]);
  std::vector<char*> cargs;
  for (auto& a : args) cargs.push_back(const_cast<char*>(a.c_str()));
  cargs.push_back(nullptr);
  fprintf(stderr, "[lux] multi-GPU: exec torchrun x%d -m %s\n", ngpus,
          module);
  execvp("python3", cargs.data());
  perror("execvp python3");
  return 127;
}

}  // namespace lux
