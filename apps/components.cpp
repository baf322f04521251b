// Native Connected Components driver (single GPU) — reference parity:
// /root/reference/components/components.cc (max-label propagation).
#include <chrono>
#include <cstdio>

#include "../src/runtime/single_gpu.h"
#include "app_common.h"

using namespace lux;

int main(int argc, char** argv) {
  AppArgs a = parse_input_args(argc, argv);
  if (a.num_gpu > 1) {
    fprintf(stderr,
            "[lux] multi-GPU runs use the RCCL engine: torchrun "
            "--nproc-per-node %d -m lux_amd.apps.cc ...\n", a.num_gpu);
    return 2;
  }
  HostCSC g;
  if (!load_graph(a, &g, false)) return 1;
  print_memory_estimate(g.nv, g.ne, false, 1);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  size_t arena_bytes = 8ull * g.nv + 4ull * g.ne
                       + 16ull * (g.nv + 1) + 4ull * g.ne
                       + 12ull * g.nv + (64ull << 20)
                       + 8ull * g.nv + 3ull * frontier_bytes(g.nv)
                       + 8ull * (g.ne / 8192 + g.nv / 16);
  DeviceArena arena(arena_bytes);
  DeviceGraph dg = DeviceGraph::upload(g, arena, s);
  SingleGpuPush engine(dg, /*is_min=*/false, 0, arena, s, a.verbose);

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
