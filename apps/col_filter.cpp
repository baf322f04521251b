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
                                  a.num_iter, a.als, a.dump, a.users);
  }
  if (a.num_gpu > 1 && getenv("LUX_TORCHRUN"))
    // escape hatch: the torchrun RCCL engine (same CLI, Python driver)
    return exec_torchrun_app("lux_amd.apps.cf", a.num_gpu, argc, argv);
  if (a.num_gpu > 1 || getenv("LUX_NATIVE_MULTI")) {
    // native fork+exec + RCCL engine, one worker per GPU
    return run_multi_workers(a.num_gpu, argc, argv);
  }
  int sscale = 0;
  long long snu = 0, sni = 0, sne = 0;
  int skind = parse_synthetic(a.synthetic, &sscale, &snu, &sni, &sne);
  if (skind == 1) {
    fprintf(stderr, "[lux] col_filter needs a weighted graph: use "
                    "-synthetic bipartite:NU:NI:NE\n");
    return 1;
  }
  HostCSC g;
  uint64_t NV = 0, NE = 0;
  if (skind == 2) {
    NV = (uint64_t)(snu + sni);
    NE = (uint64_t)sne;
  } else {
    if (!load_graph(a, &g, true)) return 1;
    NV = g.nv;
    NE = g.ne;
  }
  print_memory_estimate((V_ID)NV, NE, true, a.k);

  hipStream_t s;
  LUX_OK(hipStreamCreate(&s));
  if (a.als && a.k > 64) {
    fprintf(stderr, "[lux] -als covers K <= 64 (MFMA tile grid)\n");
    return 1;
  }
  size_t arena_bytes = 8ull * NV + 8ull * NE              // graph + weights
                       + 8ull * (NV + 1)
                       + 12ull * NV + (64ull << 20)
                       + 8ull * NV * a.k                  // old/new vectors
                       + 8ull * (NE / 8192 + NV / 16)
                       + (a.als ? 4ull * NV +             // hub slot map
                            (17ull << 20) * (NE / 2048 / 1024 + 1) : 0);
  DeviceArena arena(arena_bytes);
  DeviceGraph dg;
  if (skind) {
    build_synthetic_device(a, arena, s, &dg);
  } else {
    dg = DeviceGraph::upload(g, arena, s);
  }
  // a.users (from -users or the -synthetic bipartite spec) enables true
  // Gauss-Seidel ALS alternation; 0 falls back to simultaneous Jacobi
  SingleGpuCF engine(dg, a.k, arena, s, a.als, a.users);

  auto t0 = std::chrono::steady_clock::now();
  engine.iterate(a.num_iter);
  auto t1 = std::chrono::steady_clock::now();
  double secs = std::chrono::duration<double>(t1 - t0).count();
  printf("ELAPSED TIME = %7.7f s\n", secs);
  if (a.dump)
    dump_state(a.dump, engine.vectors(), 0, (uint32_t)a.k, dg.nv, (uint64_t)a.num_iter);
  printf("[lux] %.3f GTEPS (%d sweeps, rank %d)\n",
         double(dg.ne) * a.num_iter / secs / 1e9, a.num_iter, a.k);
  return 0;
}
