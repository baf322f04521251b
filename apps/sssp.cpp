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
  HostCSC g;
  if (!load_graph(a, &g, false)) return 1;
  print_memory_estimate(g.nv, g.ne, false, 1);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  size_t arena_bytes = 8ull * g.nv + 4ull * g.ne          // graph
                       + 16ull * (g.nv + 1) + 4ull * g.ne // csrs
                       + 12ull * g.nv + (64ull << 20)     // bins + slack
                       + 8ull * g.nv                      // labels+snapshot
                       + 3ull * frontier_bytes(g.nv)
                       + 8ull * (g.ne / 8192 + g.nv / 16)
                       + BlockedPull::arena_bytes(g.nv, g.nv, g.ne)
                       + g.nv / 2;  // BFS bits
  DeviceArena arena(arena_bytes);
  DeviceGraph dg = DeviceGraph::upload(g, arena, s);
  SingleGpuPush engine(dg, /*is_min=*/true, a.start, arena, s, a.verbose);

  auto t0 = std::chrono::steady_clock::now();
  int iters = engine.run();
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  printf("ELAPSED TIME = %7.7f s\n", secs);
  if (a.dump)
    dump_state(a.dump, engine.labels(), 1, 1, g.nv, (uint64_t)a.num_iter);
  printf("[lux] converged in %d iterations, %.3f GTEPS\n", iters,
         double(g.ne) / secs / 1e9);
  if (a.check) {
    uint64_t mistakes = engine.check();
    printf("[%s] %llu mistakes\n", mistakes == 0 ? "PASS" : "FAIL",
           (unsigned long long)mistakes);
    return mistakes == 0 ? 0 : 1;
  }
  return 0;
}
