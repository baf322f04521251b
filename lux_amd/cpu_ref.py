"""CPU reference engines (whole-graph and per-partition drivers).

These wrap the native golden implementations (src/core/engines.cpp) and add
the same iterate/exchange structure the GPU engines use, so the distributed
exchange layer can be exercised on CPU (gloo) with identical results to the
single-process path.
"""
import numpy as np

from . import _native as nat
from .graph import Graph

INF = np.uint32(0xFFFFFFFF)


def pagerank(g: Graph, iters: int) -> np.ndarray:
    """Whole-graph PageRank; returns rank/out_degree per vertex (the
    reference's stored form, pagerank_gpu.cu:97-100)."""
    return nat.pagerank_cpu(g.nv, g.ne, g.col_end, g.src, iters)


def sssp(g: Graph, source: int):
    return nat.sssp_cpu(g.nv, g.ne, g.col_end, g.src, source)


def cc(g: Graph):
    return nat.cc_cpu(g.nv, g.ne, g.col_end, g.src)


def cf(g: Graph, K: int, iters: int) -> np.ndarray:
    return nat.cf_cpu(g.nv, g.ne, g.col_end, g.src, g.weight, K, iters)


def cf_loss(g: Graph, K: int, vec: np.ndarray) -> float:
    return nat.cf_loss(g.nv, g.ne, g.col_end, g.src, g.weight, K, vec)


def sssp_check(g: Graph, label: np.ndarray) -> int:
    return nat.sssp_check(g.nv, g.ne, g.col_end, g.src, label)


def cc_check(g: Graph, label: np.ndarray) -> int:
    return nat.cc_check(g.nv, g.ne, g.col_end, g.src, label)


# ---- partitioned iteration (the distributed-CPU compute step) ----

def pagerank_partitioned(g: Graph, nparts: int, iters: int) -> np.ndarray:
    """Multi-partition PageRank in one process — validates that the
    partitioned iteration composes to the whole-graph result."""
    part = g.partition(nparts)
    deg = g.out_degrees()
    old = nat.pagerank_init(g.nv, deg)
    new = np.empty_like(old)
    for _ in range(iters):
        for p in range(nparts):
            if part.verts(p) == 0:
                continue
            rl, rr, cl, ce, src, _ = g.slice(part, p)
            nat.pagerank_iter_part(g.nv, rl, rr, cl, ce, src, deg, old,
                                   new[rl:rr + 1])
        old, new = new, old
    return old


def sssp_partitioned(g: Graph, nparts: int, source: int):
    part = g.partition(nparts)
    old = np.full(g.nv, INF, np.uint32)
    old[source] = 0
    new = np.empty_like(old)
    iters = 0
    while True:
        changed = 0
        for p in range(nparts):
            if part.verts(p) == 0:
                continue
            rl, rr, cl, ce, src, _ = g.slice(part, p)
            changed += nat.sssp_iter_part(rl, rr, cl, ce, src, old,
                                          new[rl:rr + 1])
        iters += 1
        old, new = new, old
        if changed == 0:
            break
    return old, iters


def cf_als(g: Graph, K: int, iters: int, lam: float = 0.001,
           init: "np.ndarray | None" = None,
           n_users: "int | None" = None) -> np.ndarray:
    """Plain-numpy ALS reference: per sweep, for each vertex with in-edges,
    solve (S^T S + lam I) d = S^T w exactly (S = src vectors of the
    in-edges). Vertices with no in-edges keep their old vector. With
    n_users given (bipartite user/item boundary) the sweep ALTERNATES:
    user rows [0, n_users) solve against the old item factors, then item
    rows solve against the UPDATED users — true Gauss-Seidel ALS, matching
    CFALSEngine. Without it, every row solves against old values
    (simultaneous Jacobi). Solved in float64 here, compared with tolerance
    against the fp32 GPU Cholesky path (src/gpu/cf_als.hip)."""
    import math as _math
    if init is not None:
        vec = np.array(init, dtype=np.float32).reshape(g.nv, K).copy()
    else:
        vec = np.full((g.nv, K), _math.sqrt(1.0 / K), dtype=np.float32)
    eye = lam * np.eye(K, dtype=np.float64)

    def _solve_rows(vec, new, lo, hi):
        b = 0 if lo == 0 else int(g.col_end[lo - 1])
        for v in range(lo, hi):
            e = int(g.col_end[v])
            if e > b:
                S = vec[g.src[b:e]].astype(np.float64)
                w = g.weight[b:e].astype(np.float64)
                G = S.T @ S + eye
                new[v] = np.linalg.solve(G, S.T @ w).astype(np.float32)
            b = e

    for _ in range(iters):
        if n_users is None:
            new = vec.copy()
            _solve_rows(vec, new, 0, g.nv)
            vec = new
        else:
            # in-place is exact per phase: a bipartite row only reads the
            # other side's factors
            _solve_rows(vec, vec, 0, n_users)
            _solve_rows(vec, vec, n_users, g.nv)
    return vec
