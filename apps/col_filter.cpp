// Native collaborative-filtering driver (single GPU) — reference parity:
// /root/reference/col_filter/colfilter.cc (weighted graph, -ni SGD sweeps,
// latent rank -k; reference K=20, our default 64).
#include <chrono>
#include <cstdio>

#include "../src/runtime/multi_gpu.h"
#include "../src/runtime/single_gpu.h"
#include "app_common.h"

using namespace lux;

int main(int argc, char** argv) {
  AppArgs a = parse_input_args(argc, argv);
  if (const char* mr = getenv("LUX_MULTI_RANK")) {
    // re-exec'd native multi-GPU worker (SGD or MFMA ALS sweeps)
    HostCSC g;
    if (!load_graph(a, &g, true)) return 1;
    if (a.als && a.k > 64) {
      fprintf(stderr, "[lux] -als covers K <= 64 (MFMA tile grid)\n");
      return 1;
    }
    return col_filter_multi_child(g, atoi(mr),
                                  atoi(getenv("LUX_MULTI_WORLD")),
                                  getenv("LUX_MULTI_IDFILE"), a.k,
                                  a.num_iter, a.als, a.dump);
  }
  if (a.num_gpu > 1 && getenv("LUX_TORCHRUN"))
    // escape hatch: the torchrun RCCL engine (same CLI, Python driver)
    return exec_torchrun_app("lux_amd.apps.cf", a.num_gpu, argc, argv);
  if (a.num_gpu > 1 || getenv("LUX_NATIVE_MULTI")) {
    // native fork+exec + RCCL engine, one worker per GPU
    return run_multi_workers(a.num_gpu, argc, argv);
  }
  HostCSC g;
  if (!load_graph(a, &g, true)) return 1;
  print_memory_estimate(g.nv, g.ne, true, a.k);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  if (a.als && a.k > 64) {
    fprintf(stderr, "[lux] -als covers K <= 64 (MFMA tile grid)\n");
    return 1;
  }
  size_t arena_bytes = 8ull * g.nv + 8ull * g.ne          // graph + weights
                       + 8ull * (g.nv + 1)
                       + 12ull * g.nv + (64ull << 20)
                       + 8ull * (uint64_t)g.nv * a.k      // old/new vectors
                       + 8ull * (g.ne / 8192 + g.nv / 16)
                       + (a.als ? 4ull * g.nv +           // hub slot map
                            (17ull << 20) * (g.ne / 2048 / 1024 + 1) : 0);
  DeviceArena arena(arena_bytes);
  DeviceGraph dg = DeviceGraph::upload(g, arena, s);
  SingleGpuCF engine(dg, a.k, arena, s, a.als);

  auto t0 = std::chrono::steady_clock::now();
  engine.iterate(a.num_iter);
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  printf("ELAPSED TIME = %7.7f s\n", secs);
  if (a.dump)
    dump_state(a.dump, engine.vectors(), 0, (uint32_t)a.k, g.nv, (uint64_t)a.num_iter);
  printf("[lux] %.3f GTEPS (%d sweeps, rank %d)\n",
         double(g.ne) * a.num_iter / secs / 1e9, a.num_iter, a.k);
  return 0;
}
