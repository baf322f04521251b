// Shared CLI parsing for the native app binaries — flag surface parity with
// the reference drivers (sssp/sssp.cc:148-180, README.md:42-54): -ng/-ll:gpu,
// -ni, -file, -start, -verbose/-v, -check/-c; -ll:fsize/-ll:zsize and other
// Legion flags are accepted and ignored. Extra (ours): -synthetic spec, -k.
//
// These binaries run the single-GPU native runtime (src/runtime/); the
// multi-GPU path is `torchrun ... python -m lux_amd.apps.<app>` (one process
// per GPU over RCCL). -ng > 1 here prints a pointer to that path.
#pragma once
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "../src/runtime/single_gpu.h"
#include "lux/graph.h"

struct AppArgs {
  int num_gpu = 1;
  int num_iter = 10;
  const char* file = nullptr;
  lux::V_ID start = 0;
  bool verbose = false;
  bool check = false;
  int k = 64;
  const char* synthetic = nullptr;
  const char* dump = nullptr;  // write result vertex state (LUXS format,
                               // lux_amd/checkpoint.py-compatible)
  bool als = false;            // col_filter: exact MFMA ALS optimizer
  bool labelprop = false;      // components: reference-parity label prop
  lux::V_ID users = 0;         // col_filter -als: bipartite user/item
                               // boundary for Gauss-Seidel alternation
                               // (auto-set from -synthetic bipartite:...)
};

inline AppArgs parse_input_args(int argc, char** argv) {
  AppArgs a;
  for (int i = 1; i < argc; i++) {
    std::string f = argv[i];
    auto next = [&]() { return argv[++i]; };
    if (f == "-ng" || f == "-ll:gpu") a.num_gpu = atoi(next());
    else if (f == "-ni") a.num_iter = atoi(next());
    else if (f == "-file") a.file = next();
    else if (f == "-start") a.start = (lux::V_ID)atoll(next());
    else if (f == "-verbose" || f == "-v") a.verbose = true;
    else if (f == "-check" || f == "-c") a.check = true;
    else if (f == "-k") a.k = atoi(next());
    else if (f == "-synthetic") a.synthetic = next();
    else if (f == "-dump") a.dump = next();
    else if (f == "-als") a.als = true;
    else if (f == "-users") a.users = (lux::V_ID)atoll(next());
    else if (f == "-labelprop") a.labelprop = true;
    else if (f.rfind("-ll:", 0) == 0 || f.rfind("-lg:", 0) == 0) {
      if (i + 1 < argc && argv[i + 1][0] != '-') i++;  // value-flag: skip
    } else {
      fprintf(stderr, "warning: ignoring unknown flag %s\n", f.c_str());
    }
  }
  if (a.users == 0 && a.synthetic) {  // boundary is implicit in the spec
    long long nu, ni, ne;
    if (sscanf(a.synthetic, "bipartite:%lld:%lld:%lld", &nu, &ni, &ne) == 3)
      a.users = (lux::V_ID)nu;
  }
  return a;
}

// Vertex-state dump: u32 magic 'LUXS' | u32 dtype (0=f32,1=u32) | u32 K |
// u32 nv | u64 iteration | payload (lux_amd/checkpoint.py format).
inline void dump_state(const char* path, const void* dev_ptr, int dtype,
                       uint32_t k, uint32_t nv, uint64_t iter) {
  std::vector<char> host((size_t)nv * k * 4);
  LUX_OK(hipMemcpy(host.data(), dev_ptr, host.size(),
                   hipMemcpyDeviceToHost));
  FILE* f = fopen(path, "wb");
  if (!f) { perror(path); return; }
  uint32_t hdr[4] = {0x5358554Cu, (uint32_t)dtype, k, nv};
  fwrite(hdr, 4, 4, f);
  fwrite(&iter, 8, 1, f);
  fwrite(host.data(), 1, host.size(), f);
  fclose(f);
  printf("[lux] wrote %s (nv=%u k=%u)\n", path, nv, k);
}

// Parse -synthetic without building: returns 0 (not synthetic / unknown),
// 1 (rmat:S:NE), 2 (bipartite:NU:NI:NE); fills the shape outputs.
inline int parse_synthetic(const char* spec, int* scale, long long* nu,
                           long long* ni, long long* ne) {
  if (!spec) return 0;
  if (sscanf(spec, "rmat:%d:%lld", scale, ne) == 2) return 1;
  if (sscanf(spec, "bipartite:%lld:%lld:%lld", nu, ni, ne) == 3) return 2;
  return 0;
}

// Device-side synthetic build for the single-GPU drivers (the CPU
// generator at RMAT-27 takes minutes; the GPU one milliseconds). Returns
// true and fills *dg when -synthetic was used.
inline bool build_synthetic_device(const AppArgs& a, lux::DeviceArena& arena,
                                   hipStream_t s, lux::DeviceGraph* dg) {
  int scale = 0;
  long long nu = 0, ni = 0, ne = 0;
  int kind = parse_synthetic(a.synthetic, &scale, &nu, &ni, &ne);
  if (kind == 1) {
    *dg = lux::DeviceGraph::rmat(scale, (lux::E_ID)ne, 1, arena, s);
    return true;
  }
  if (kind == 2) {
    *dg = lux::DeviceGraph::bipartite((lux::V_ID)nu, (lux::V_ID)ni,
                                      (lux::E_ID)ne, 1, arena, s);
    return true;
  }
  return false;
}

inline bool load_graph(const AppArgs& a, lux::HostCSC* g, bool weighted) {
  if (a.file) return lux::lux_read(a.file, g, weighted);
  if (a.synthetic) {
    int scale;
    long long ne;
    if (sscanf(a.synthetic, "rmat:%d:%lld", &scale, &ne) == 2) {
      *g = lux::rmat_csc_cpu(scale, (lux::E_ID)ne, 1);
      return true;
    }
    long long nu, ni;
    if (sscanf(a.synthetic, "bipartite:%lld:%lld:%lld", &nu, &ni, &ne) == 3) {
      *g = lux::bipartite_csc_cpu((lux::V_ID)nu, (lux::V_ID)ni,
                                  (lux::E_ID)ne, 1);
      return true;
    }
  }
  fprintf(stderr, "usage: need -file graph.lux or -synthetic rmat:S:NE\n");
  return false;
}

inline void print_memory_estimate(lux::V_ID nv, lux::E_ID ne, bool weighted,
                                  int k) {
  double fb = double(ne) * (weighted ? 8 : 4) + 48.0 * nv +
              8.0 * k * nv;
  printf("[lux] estimated FB usage: %.0f MB (of 294912 MB HBM3E)\n",
         fb / (1 << 20));
}
