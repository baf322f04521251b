// Native SSSP driver (single GPU) — reference parity: /root/reference/sssp/sssp.cc
// (unweighted hop distances, push model, -start source, -check oracle).
#include <chrono>
#include <cstdio>

#include "../src/runtime/multi_gpu.h"
#include "../src/runtime/single_gpu.h"
#include "app_common.h"

using namespace lux;

int main(int argc, char** argv) {
  AppArgs a = parse_input_args(argc, argv);
  if (const char* mr = getenv("LUX_MULTI_RANK")) {
    // re-exec'd native multi-GPU push worker (meta-record exchange)
    HostCSC g;
    if (!load_graph(a, &g, false)) return 1;
    return push_multi_child(g, atoi(mr), atoi(getenv("LUX_MULTI_WORLD")),
                            /*is_min=*/true, a.start,
                            getenv("LUX_MULTI_IDFILE"), a.check, a.dump,
                            a.verbose);
  }
  if (a.num_gpu > 1 && getenv("LUX_TORCHRUN"))
    // escape hatch: the torchrun RCCL engine (same CLI, Python driver)
    return exec_torchrun_app("lux_amd.apps.sssp", a.num_gpu, argc, argv);
  if (a.num_gpu > 1 || getenv("LUX_NATIVE_MULTI")) {
    // native fork+exec + RCCL push engine, one worker per GPU
    return run_multi_workers(a.num_gpu, argc, argv);
  }
  int sscale = 0;
  long long snu = 0, sni = 0, sne = 0;
  int skind = parse_synthetic(a.synthetic, &sscale, &snu, &sni, &sne);
  HostCSC g;
  uint64_t NV, NE;
  if (skind == 1) {
    NV = 1ull << sscale;
    NE = (uint64_t)sne;
  } else if (skind == 2) {
    NV = (uint64_t)(snu + sni);
    NE = (uint64_t)sne;
  } else {
    skind = 0;
  }
  if (!skind) {
    if (!load_graph(a, &g, false)) return 1;
    NV = g.nv;
    NE = g.ne;
  }
  print_memory_estimate((V_ID)NV, NE, false, 1);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  size_t arena_bytes = 8ull * NV + 4ull * NE              // graph
                       + 16ull * (NV + 1) + 4ull * NE     // csrs
                       + 12ull * NV + (64ull << 20)       // bins + slack
                       + 8ull * NV                        // labels+snapshot
                       + 3ull * frontier_bytes((V_ID)NV)
                       + 8ull * (NE / 8192 + NV / 16)
                       + BlockedPull::arena_bytes((V_ID)NV, (V_ID)NV, NE)
                       + NV / 2;  // BFS bits
  DeviceArena arena(arena_bytes);
  DeviceGraph dg;
  if (skind) {
    // synthetic graphs build on-device (CPU gen at RMAT-27 takes minutes)
    build_synthetic_device(a, arena, s, &dg);
  } else {
    dg = DeviceGraph::upload(g, arena, s);
  }
  SingleGpuPush engine(dg, /*is_min=*/true, a.start, arena, s, a.verbose);

  auto t0 = std::chrono::steady_clock::now();
  int iters = engine.run();
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  printf("ELAPSED TIME = %7.7f s\n", secs);
  if (a.dump)
    dump_state(a.dump, engine.labels(), 1, 1, dg.nv, (uint64_t)a.num_iter);
  printf("[lux] converged in %d iterations, %.3f GTEPS\n", iters,
         double(dg.ne) / secs / 1e9);
  if (a.check) {
    uint64_t mistakes = engine.check();
    printf("[%s] %llu mistakes\n", mistakes == 0 ? "PASS" : "FAIL",
           (unsigned long long)mistakes);
    return mistakes == 0 ? 0 : 1;
  }
  return 0;
}
