// Text edge list -> .lux CSC converter. Reference parity:
// /root/reference/tools/converter.cc (flags -nv -ne -input -output; input
// lines "src dst" or "src dst weight" with -weighted). Differences, per
// SURVEY.md §7 "fix" list: uses a counting sort by dst instead of a
// comparison sort, and does NOT append the trailing unread degree block
// (no reference loader reads it; our reader tolerates files that have it).
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "lux/graph.h"

using namespace lux;

int main(int argc, char** argv) {
  long long nv = -1, ne = -1;
  const char *input = nullptr, *output = nullptr;
  bool weighted = false;
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "-nv")) nv = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-ne")) ne = atoll(argv[++i]);
    else if (!strcmp(argv[i], "-input")) input = argv[++i];
    else if (!strcmp(argv[i], "-output")) output = argv[++i];
    else if (!strcmp(argv[i], "-weighted")) weighted = true;
  }
  if (nv < 0 || ne < 0 || !input || !output) {
    fprintf(stderr,
            "usage: converter -nv NV -ne NE -input edges.txt -output g.lux "
            "[-weighted]\n");
    return 1;
  }
  FILE* f = fopen(input, "r");
  if (!f) {
    fprintf(stderr, "cannot open %s\n", input);
    return 1;
  }
  std::vector<V_ID> src(ne), dst(ne);
  std::vector<WeightType> w;
  if (weighted) w.resize(ne);
  for (long long e = 0; e < ne; e++) {
    unsigned u, v;
    int wt = 0;
    int got = weighted ? fscanf(f, "%u %u %d", &u, &v, &wt)
                       : fscanf(f, "%u %u", &u, &v);
    if (got < (weighted ? 3 : 2)) {
      fprintf(stderr, "short read at edge %lld\n", e);
      return 1;
    }
    src[e] = u;
    dst[e] = v;
    if (weighted) w[e] = wt;
  }
  fclose(f);
  HostCSC g = edges_to_csc((V_ID)nv, src, dst, weighted ? &w : nullptr);
  if (!lux_write(output, g)) {
    fprintf(stderr, "write failed: %s\n", output);
    return 1;
  }
  printf("wrote %s: nv=%lld ne=%lld%s\n", output, nv, ne,
         weighted ? " (weighted)" : "");
  return 0;
}
