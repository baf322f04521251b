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

// Launcher: fork + EXEC one child per GPU (a plain fork would inherit the
// parent's process state; exec gives each rank a fresh runtime). The
// parent itself never touches HIP or RCCL.
int run_pagerank_multi(int ngpus, int argc, char** argv) {
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
