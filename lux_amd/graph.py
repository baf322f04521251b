"""Host-side graph container: CSC storage, .lux IO, synthetic generators,
edge-balanced partitioning.

Mirrors the reference's Graph data model (core/pull_model.inl:29-191,
core/push_model.inl:301-509) without Legion: a Graph is plain numpy CSC plus
the partition table; GPU-resident structures are built per-rank by the
engines (lux_amd/engine.py) from a partition slice of this object.
"""
import numpy as np

from . import _native as nat


class Partition:
    """Edge-balanced contiguous vertex ranges (one per GPU/rank)."""

    def __init__(self, nv, ne, col_end, nparts):
        self.nparts = nparts
        rl, rr, cl, cr = nat.partition(nv, ne, col_end, nparts)
        self.row_left, self.row_right = rl, rr
        self.col_left, self.col_right = cl, cr

    def verts(self, p):
        if self.row_left[p] > self.row_right[p]:
            return 0
        return int(self.row_right[p]) - int(self.row_left[p]) + 1

    def edges(self, p):
        return int(self.col_right[p]) - int(self.col_left[p])


class Graph:
    """CSC graph: col_end (u64[nv], END offsets per the .lux convention),
    src (u32[ne], grouped by dst), optional weight (i32[ne])."""

    def __init__(self, nv, ne, col_end, src, weight=None):
        self.nv = int(nv)
        self.ne = int(ne)
        self.col_end = col_end
        self.src = src
        self.weight = weight

    # ---- constructors ----
    @classmethod
    def load(cls, path, want_weights=False):
        nv, ne, col_end, src, weight = nat.io_read(path, want_weights)
        return cls(nv, ne, col_end, src, weight)

    @classmethod
    def rmat(cls, scale, ne, seed=1, sym=False):
        npairs = ne // 2 if sym else ne
        src, dst = nat.rmat_edges(seed, scale, npairs)
        if sym:  # undirected: both directions (matches DeviceCSC.rmat sym)
            src, dst = (np.concatenate([src, dst]),
                        np.concatenate([dst, src]))
        col_end, csrc, _ = nat.edges_to_csc(1 << scale, src, dst)
        return cls(1 << scale, len(src), col_end, csrc)

    @classmethod
    def rmat_folded(cls, nv, ne, seed=1, sym=False):
        """RMAT skew with a non-power-of-two nv (Twitter-shaped synthetics)."""
        scale = 0
        while (1 << scale) < nv:
            scale += 1
        npairs = ne // 2 if sym else ne
        src, dst = nat.rmat_edges_folded(seed, scale, nv, npairs)
        if sym:
            src, dst = (np.concatenate([src, dst]),
                        np.concatenate([dst, src]))
        col_end, csrc, _ = nat.edges_to_csc(nv, src, dst)
        return cls(nv, len(src), col_end, csrc)

    @classmethod
    def bipartite(cls, n_users, n_items, ne, seed=1):
        src, dst, w = nat.bipartite_edges(seed, n_users, n_items, ne)
        col_end, csrc, cw = nat.edges_to_csc(n_users + n_items, src, dst, w)
        return cls(n_users + n_items, ne, col_end, csrc, cw)

    @classmethod
    def from_edges(cls, nv, src, dst, weight=None):
        src = np.asarray(src, np.uint32)
        dst = np.asarray(dst, np.uint32)
        w = np.asarray(weight, np.int32) if weight is not None else None
        col_end, csrc, cw = nat.edges_to_csc(nv, src, dst, w)
        return cls(nv, len(src), col_end, csrc, cw)

    # ---- IO ----
    def save(self, path):
        nat.io_write(path, self.nv, self.ne, self.col_end, self.src,
                     self.weight)

    # ---- structure ----
    def row_begin(self, v):
        return 0 if v == 0 else int(self.col_end[v - 1])

    def row_end(self, v):
        return int(self.col_end[v])

    def partition(self, nparts):
        return Partition(self.nv, self.ne, self.col_end, nparts)

    def out_degrees(self):
        return nat.out_degrees(self.nv, self.src)

    def slice(self, part, p):
        """Partition slice views: (row_left, row_right, col_left,
        col_end_slice, src_slice, weight_slice)."""
        rl, rr = int(part.row_left[p]), int(part.row_right[p])
        cl, cr = int(part.col_left[p]), int(part.col_right[p])
        w = self.weight[cl:cr] if self.weight is not None else None
        return rl, rr, cl, self.col_end[rl:rr + 1], self.src[cl:cr], w

    def memory_estimate_bytes(self, nparts=1, k=1):
        """Startup memory-requirement estimate per GPU, the MI355X analog of
        the reference's FB/ZC printout (pagerank.cc:60-85, sssp.cc:59-90)."""
        per_edge = 4 + (4 if self.weight is not None else 0)
        state = 4 * k
        fb = (self.ne * per_edge) // max(nparts, 1) + 8 * (self.nv // max(nparts, 1)) \
            + 2 * state * self.nv
        return fb
