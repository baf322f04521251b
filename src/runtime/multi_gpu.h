// Native multi-GPU runtime: fork one child process per GPU, RCCL over
// xGMI for the exchange (the BASELINE north star: C++/HIP + RCCL with no
// Python in the hot path). `bin/pagerank -ll:gpu N` runs this engine
// natively; the push/CF apps bridge to the torchrun engine (same RCCL
// collectives, Python driver) via exec_torchrun_app.
#pragma once
#include "single_gpu.h"

namespace lux {

// Launcher: fork+exec one re-invocation of this binary per GPU (children
// detected via LUX_MULTI_RANK env). The parent never initialises HIP.
int run_pagerank_multi(int ngpus, int argc, char** argv);

// Child entry: device `rank`, RCCL communicator joined via the id file
// rank 0 writes; distributed pull PageRank over an edge-balanced
// partition of g.
int pagerank_multi_child(const HostCSC& g, int rank, int ngpus,
                         const char* idfile, int iters, bool verbose,
                         const char* dump);

// Native multi-GPU workers for the other apps (same launcher: the app
// main re-invokes these under LUX_MULTI_RANK).
int components_multi_child(const HostCSC& g, int rank, int ngpus,
                           const char* idfile, bool check, const char* dump,
                           bool verbose);
// n_users > 0 (bipartite boundary) makes each ALS sweep alternate:
// user half-sweep, publish, item half-sweep against updated users.
int col_filter_multi_child(const HostCSC& g, int rank, int ngpus,
                           const char* idfile, int K, int iters, bool als,
                           const char* dump, lux::V_ID n_users = 0);
int push_multi_child(const HostCSC& g, int rank, int ngpus, bool is_min,
                     lux::V_ID source, const char* idfile, bool check,
                     const char* dump, bool verbose);
// Generic launcher (same as run_pagerank_multi's body): fork+exec one
// re-invocation per GPU.
int run_multi_workers(int ngpus, int argc, char** argv);

// Replace this process with `torchrun --nproc-per-node N -m <module>
// <original args>` (one rank per GPU over the same RCCL exchange layer,
// Python driver). Returns only on exec failure.
int exec_torchrun_app(const char* module, int ngpus, int argc, char** argv);

}  // namespace lux
