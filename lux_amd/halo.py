"""Halo (in_vtxs) exchange for the pull engines.

The reference gathers, per iteration, only the source vertices its
partition actually reads — the `in_vtxs` halo list built at init
(pagerank_gpu.cu:229-241) — instead of every peer's whole slice. Here the
same idea runs over RCCL: each rank marks the sources appearing in its
edge slice, exchanges STATIC per-peer index lists once at init, and per
iteration ships only the requested values (pack -> batched p2p ->
scatter into the replicated array at exactly the halo positions).

Pays when the union of halo lists is smaller than the full slice
exchange — e.g. mesh/road-shaped partitions whose in-edges are mostly
local. On RMAT graphs connectivity is dense (halo ~ nv) and the engines
keep the plain slice all-gather; `HaloExchange.worth_it` decides (or
LUX_HALO=1/0 forces).

Correctness contract: after publish(), the replicated array is current
at every position any pull sweep of any rank reads (its edge sources +
its own slice). Positions outside the halo union may be stale — callers
that hand the replicated array to users do one full all-gather first
(LabelPullEngine.labels, PushEngine.final_labels).
"""
import torch

from . import dist as dx


class _HaloWork:
    def __init__(self, reqs, scatters):
        self.reqs = reqs
        self.scatters = scatters

    def wait(self):
        for r in self.reqs:
            r.wait()
        # stream-ordered after the recvs (NCCL); host-ordered for gloo
        for full, left, idx, buf in self.scatters:
            full[left + idx] = buf
        return True


class HaloExchange:
    def __init__(self, nv, row_left_all, verts_all, my_part, col,
                 elems_per_vertex=1):
        """col: this rank's edge-source ids (u32-as-int32 tensor, device
        or CPU). elems_per_vertex: K for CF-style vector properties."""
        self.nv = nv
        self.row_left_all = row_left_all
        self.verts_all = verts_all
        self.p = my_part
        self.K = elems_per_vertex
        ws = len(verts_all)
        device = col.device
        assert nv < (1 << 31), "halo lists index with signed int64 views"
        mask = torch.zeros(nv, dtype=torch.bool, device=device)
        mask[col.long()] = True
        # my needed indices per peer, LOCAL to the peer's range
        self.recv_idx = []
        for q in range(ws):
            if q == my_part or verts_all[q] == 0:
                self.recv_idx.append(torch.zeros(0, dtype=torch.long,
                                                 device=device))
                continue
            rng = mask.narrow(0, row_left_all[q], verts_all[q])
            self.recv_idx.append(rng.nonzero(as_tuple=False).view(-1))
        recv_counts = [int(t.numel()) for t in self.recv_idx]

        if not dx.initialized() or dx.world_size() == 1:
            # single process: nothing to send; publish copies own slice
            self.send_idx = [torch.zeros(0, dtype=torch.long,
                                         device=device)] * ws
            self.send_counts = [0] * ws
            self.recv_counts = recv_counts
            self._cnt = None
            self._bufs_ready = False
            return

        # exchange counts only (the cheap part: a ws*ws int matrix;
        # tensors live on `device` so the collective works on NCCL too).
        # The index lists themselves are exchanged LAZILY — only after
        # worth_it() said yes on every rank (the decision is computed
        # from the full matrix, so it is globally identical; per-rank
        # decisions could deadlock the list exchange).
        cnt = torch.zeros(ws * ws, dtype=torch.int64, device=device)
        mine = torch.tensor(recv_counts, dtype=torch.int64, device=device)
        dx.all_gather_slices(cnt, mine, [ws] * ws,
                             [q * ws for q in range(ws)], my_index=my_part)
        self._cnt = cnt.view(ws, ws).cpu()  # cnt[p][q] = p wants from q
        self.send_counts = [int(self._cnt[q, my_part]) for q in range(ws)]
        self.recv_counts = recv_counts
        self.send_idx = None  # exchanged by _exchange_lists()
        self._bufs_ready = False

    def _exchange_lists(self):
        """Collective: ship my per-peer index lists; receive the lists
        peers will be served from. Call only after worth_it() (globally
        consistent) returned True on every rank."""
        if self.send_idx is not None:
            return
        import torch.distributed as dist
        ws = len(self.verts_all)
        my_part = self.p
        device = self.recv_idx[0].device if self.recv_idx else "cpu"
        ops = []
        pend = []
        for off in range(1, ws):
            ps = (my_part + off) % ws
            pr = (my_part - off) % ws
            if self.recv_counts[ps]:
                ops.append(dist.P2POp(dist.isend,
                                      self.recv_idx[ps].contiguous(), ps))
            if self.send_counts[pr]:
                buf = torch.empty(self.send_counts[pr], dtype=torch.long,
                                  device=device)
                pend.append((pr, buf))
                ops.append(dist.P2POp(dist.irecv, buf, pr))
        if ops:
            for r in dist.batch_isend_irecv(ops):
                r.wait()
        sidx = [torch.zeros(0, dtype=torch.long, device=device)] * ws
        for pr, buf in pend:
            sidx[pr] = buf
        self.send_idx = sidx

    def total_halo(self):
        """Vertices received per iteration (the wire cost)."""
        return sum(self.recv_counts)

    def worth_it(self):
        """Enable halo only on a DECISIVE margin (< 35% of the full
        exchange): the pack/scatter indexing is not free, and power-law
        graphs whose halo is merely somewhat smaller than nv are better
        served by the contiguous slice all-gather. LUX_HALO=1/0
        forces."""
        import os
        f = os.environ.get("LUX_HALO")
        if f is not None:
            return f == "1"
        if getattr(self, "_cnt", None) is None:
            return False  # single process: nothing to exchange
        # GLOBAL decision (identical on every rank): total halo elements
        # over all rank pairs vs the total full-slice exchange
        ws = len(self.verts_all)
        halo_cost = int(self._cnt.sum()) * 2  # each element recv'd + sent
        full_cost = sum(
            (sum(self.verts_all) - self.verts_all[p]) * 2
            for p in range(ws))
        return halo_cost * 100 < full_cost * 35

    def _ensure_bufs(self, dtype, device):
        if self._bufs_ready:
            return
        K = self.K
        self.send_bufs = [
            torch.empty(self.send_counts[q] * K, dtype=dtype, device=device)
            for q in range(len(self.send_counts))]
        self.recv_bufs = [
            torch.empty(self.recv_counts[q] * K, dtype=dtype, device=device)
            for q in range(len(self.recv_counts))]
        self._bufs_ready = True

    def publish_async(self, full, my_slice):
        """Publish my slice: copy it into my window of `full` and ship
        each peer exactly the values its halo list names. Returns a
        Work-like handle; after .wait(), `full` is current at halo + own
        positions."""
        import torch.distributed as dist
        p = self.p
        ws = len(self.verts_all)
        left = self.row_left_all
        if self.verts_all[p]:
            dst = full.narrow(0, left[p] * self.K,
                              self.verts_all[p] * self.K)
            if dst.data_ptr() != my_slice.data_ptr():
                dst.copy_(my_slice)
        if not dx.initialized() or dx.world_size() == 1:
            return dx._DoneWork()
        self._exchange_lists()  # lazy, first publish (collective)
        self._ensure_bufs(full.dtype, full.device)
        K = self.K
        mv = my_slice.view(self.verts_all[p], K) if K > 1 else my_slice
        ops = []
        scatters = []
        for off in range(1, ws):
            ps = (p + off) % ws
            pr = (p - off) % ws
            if self.send_counts[ps]:
                buf = self.send_bufs[ps]
                src = mv[self.send_idx[ps]].reshape(-1) if K > 1 \
                    else my_slice[self.send_idx[ps]]
                buf.copy_(src)
                ops.append(dist.P2POp(dist.isend, buf, ps))
            if self.recv_counts[pr]:
                ops.append(dist.P2POp(dist.irecv, self.recv_bufs[pr], pr))
        reqs = dist.batch_isend_irecv(ops) if ops else []
        for q in range(ws):
            if self.recv_counts[q]:
                if K > 1:
                    fv = full.view(self.nv, K)
                    scatters.append((fv, left[q], self.recv_idx[q],
                                     self.recv_bufs[q].view(-1, K)))
                else:
                    scatters.append((full, left[q], self.recv_idx[q],
                                     self.recv_bufs[q]))
        return _HaloWork(reqs, scatters)

    def publish(self, full, my_slice):
        self.publish_async(full, my_slice).wait()
