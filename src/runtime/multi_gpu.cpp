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
      lux_gpu_pull_iter((uint64_t)s, 0, bins.n0, bins.bin0, bins.n1,
                        bins.bin1, bins.n2, bins.bin2, bins.nbig, bins.bin2v,
                        row_ptr, 0, col, old_, new_, deg, rl, init_rank);
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

// upload ONLY my slice (call after the arena exists on my device)
void multi_upload(const HostCSC& g, bool weighted, DeviceArena& arena,
                  MultiCtx* c) {
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
                           const char* dump) {
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
    LUX_OK(hipMemcpyAsync(old_, h.data(), 4ull * g.nv * K,
                          hipMemcpyHostToDevice, c.s));
  }
  int* hubidx = nullptr;
  float *gram = nullptr, *rhs = nullptr;
  if (als && c.bins.nbig) {
    hubidx = arena.alloc_n<int>(c.vp ? c.vp : 1);
    gram = arena.alloc_n<float>((uint64_t)c.bins.nbig * 64 * 64);
    rhs = arena.alloc_n<float>((uint64_t)c.bins.nbig * 64);
    std::vector<V_ID> hv(c.bins.nbig);
    LUX_OK(hipMemcpyAsync(hv.data(), c.bins.bin2v, 4ull * c.bins.nbig,
                          hipMemcpyDeviceToHost, c.s));
    LUX_OK(hipStreamSynchronize(c.s));
    std::vector<int> hidx(c.vp, -1);
    for (uint32_t i = 0; i < c.bins.nbig; i++) hidx[hv[i]] = (int)i;
    LUX_OK(hipMemcpyAsync(hubidx, hidx.data(), 4ull * c.vp,
                          hipMemcpyHostToDevice, c.s));
  }
  float* bar = arena.alloc_n<float>(1);
  multi_barrier(c, bar);
  double t0 = now_seconds();
  for (int it = 0; it < iters; it++) {
    if (c.vp) {
      if (als) {
        LUX_OK(hipMemcpyAsync(new_, old_ + (uint64_t)c.rl * K,
                              4ull * c.vp * K, hipMemcpyDeviceToDevice,
                              c.s));
        if (c.bins.nbig) {
          LUX_OK(hipMemsetAsync(gram, 0, 4ull * c.bins.nbig * 64 * 64,
                                c.s));
          LUX_OK(hipMemsetAsync(rhs, 0, 4ull * c.bins.nbig * 64, c.s));
        }
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
    // all-gather(v) of K-dim vector slices (direct pairwise over xGMI)
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
