// Native PageRank driver (single GPU) — reference parity:
// /root/reference/pagerank/pagerank.cc. Same CLI, same `ELAPSED TIME` line.
#include <chrono>
#include <cstdio>

#include "../src/runtime/multi_gpu.h"
#include "../src/runtime/single_gpu.h"
#include "app_common.h"

using namespace lux;

int main(int argc, char** argv) {
  AppArgs a = parse_input_args(argc, argv);
  if (const char* mr = getenv("LUX_MULTI_RANK")) {
    // re-exec'd worker of the native multi-GPU launcher below
    HostCSC g;
    if (!load_graph(a, &g, false)) return 1;
    return pagerank_multi_child(g, atoi(mr), atoi(getenv("LUX_MULTI_WORLD")),
                                getenv("LUX_MULTI_IDFILE"), a.num_iter,
                                a.verbose, a.dump);
  }
  if (a.num_gpu > 1 && getenv("LUX_TORCHRUN"))
    // escape hatch: the torchrun RCCL engine (same CLI, Python driver)
    return exec_torchrun_app("lux_amd.apps.pagerank", a.num_gpu, argc, argv);
  if (a.num_gpu > 1 || getenv("LUX_NATIVE_MULTI")) {
    // native fork+exec + RCCL engine, one worker process per GPU (the
    // reference's `pagerank -ll:gpu N` drop-in, README.md:42).
    // LUX_NATIVE_MULTI=1 forces this path at -ll:gpu 1 so the fork+RCCL
    // machinery is testable on a 1-GPU box.
    return run_pagerank_multi(a.num_gpu, argc, argv);
  }
  int sscale = 0;
  long long snu = 0, sni = 0, sne = 0;
  int skind = parse_synthetic(a.synthetic, &sscale, &snu, &sni, &sne);
  HostCSC g;
  uint64_t NV = 0, NE = 0;
  if (skind == 1) {
    NV = 1ull << sscale;
    NE = (uint64_t)sne;
  } else if (skind == 2) {
    NV = (uint64_t)(snu + sni);
    NE = (uint64_t)sne;
  } else {
    if (!load_graph(a, &g, false)) return 1;
    NV = g.nv;
    NE = g.ne;
  }
  print_memory_estimate((V_ID)NV, NE, false, 1);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  size_t arena_bytes = 8ull * NV + 4ull * NE            // graph
                       + 8ull * (NV + 1)                // row_ptr
                       + 12ull * NV + (64ull << 20)     // bins + slack
                       + 4ull * NV                      // degrees
                       + 8ull * NV                      // old/new
                       + 8ull * (NE / 8192 + NV / 16)
                       + BlockedPull::arena_bytes((V_ID)NV, (V_ID)NV, NE);
  DeviceArena arena(arena_bytes);
  DeviceGraph dg;
  if (skind) {
    // synthetic graphs build on-device (CPU gen at RMAT-27 takes minutes)
    build_synthetic_device(a, arena, s, &dg);
  } else {
    dg = DeviceGraph::upload(g, arena, s);
  }
  SingleGpuPagerank engine(dg, arena, s);

  auto t0 = std::chrono::steady_clock::now();
  engine.iterate(a.num_iter);
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  printf("ELAPSED TIME = %7.7f s\n", secs);
  if (a.dump)
    dump_state(a.dump, engine.ranks(), 0, 1, dg.nv, (uint64_t)a.num_iter);
  printf("[lux] %.3f GTEPS (%d iterations, %llu edges)\n",
         double(dg.ne) * a.num_iter / secs / 1e9, a.num_iter,
         (unsigned long long)dg.ne);
  if (a.verbose) {
    float first[5];
    LUX_OK(hipMemcpy(first, engine.ranks(), sizeof(first),
                     hipMemcpyDeviceToHost));
    printf("[lux] first ranks (pr/out_degree): %g %g %g %g %g\n", first[0],
           first[1], first[2], first[3], first[4]);
  }
  return 0;
}
