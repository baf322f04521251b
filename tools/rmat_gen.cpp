// Synthetic .lux graph generator (RMAT / folded-RMAT / bipartite) — the
// offline companion to the in-engine GPU generators; same deterministic
// per-edge function (src/include/lux/rmat.h), so files written here match
// what bench.py generates on-device for the same seed.
//
//   rmat_gen -kind rmat -scale 20 -ne 16000000 -seed 1 -output g.lux
//   rmat_gen -kind folded -nv 41652230 -ne 1468365182 -output t.lux
//   rmat_gen -kind bipartite -users 480189 -items 17770 -ne 200961014 -o n.lux
//   -sym: emit both directions of ne/2 generated pairs (undirected graphs —
//   the components app's proper input; matches DeviceCSC.rmat(sym=True))
#include <cstdio>
#include <cstdlib>
#include <cstring>

#include "lux/graph.h"
#include "lux/rmat.h"

using namespace lux;

int main(int argc, char** argv) {
  const char* kind = "rmat";
  const char* output = nullptr;
  int scale = 16;
  long long ne = 1 << 20, nv = -1, users = 0, items = 0;
  uint64_t seed = 1;
  bool sym = false;
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "-kind")) kind = argv[++i];
    else if (!strcmp(argv[i], "-scale")) scale = atoi(argv[++i]);
    else if (!strcmp(argv[i], "-ne")) ne = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-nv")) nv = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-users")) users = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-items")) items = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-seed")) seed = strtoull(argv[++i], 0, 10);
    else if (!strcmp(argv[i], "-sym")) sym = true;
    else if (!strcmp(argv[i], "-output") || !strcmp(argv[i], "-o"))
      output = argv[++i];
  }
  if (!output) {
    fprintf(stderr, "usage: rmat_gen -kind rmat|folded|bipartite ... -o out.lux\n");
    return 1;
  }
  HostCSC g;
  if (!strcmp(kind, "rmat")) {
    if (sym) {
      long long np = ne / 2;
      std::vector<V_ID> s(2 * np), d(2 * np);
      for (long long e = 0; e < np; e++) {
        rmat_edge(seed, e, scale, &s[e], &d[e]);
        s[np + e] = d[e];
        d[np + e] = s[e];
      }
      g = edges_to_csc((V_ID)1 << scale, s, d, nullptr);
    } else {
      g = rmat_csc_cpu(scale, (E_ID)ne, seed);
    }
  } else if (!strcmp(kind, "folded")) {
    if (nv <= 0) { fprintf(stderr, "folded needs -nv\n"); return 1; }
    int sc = 0;
    while ((1ll << sc) < nv) sc++;
    long long np = sym ? ne / 2 : ne;
    std::vector<V_ID> s(sym ? 2 * np : np), d(sym ? 2 * np : np);
    for (long long e = 0; e < np; e++)
      rmat_edge_folded(seed, e, sc, (V_ID)nv, &s[e], &d[e]);
    if (sym)
      for (long long e = 0; e < np; e++) {
        s[np + e] = d[e];
        d[np + e] = s[e];
      }
    g = edges_to_csc((V_ID)nv, s, d, nullptr);
  } else if (!strcmp(kind, "bipartite")) {
    g = bipartite_csc_cpu((V_ID)users, (V_ID)items, (E_ID)ne, seed);
  } else {
    fprintf(stderr, "unknown -kind %s\n", kind);
    return 1;
  }
  if (!lux_write(output, g)) {
    fprintf(stderr, "write failed\n");
    return 1;
  }
  printf("wrote %s: nv=%u ne=%llu%s\n", output, g.nv,
         (unsigned long long)g.ne, g.weighted() ? " (weighted)" : "");
  return 0;
}
