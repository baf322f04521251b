"""Distributed exchange layer tests on CPU (gloo, world_size 2) — validates
the all-gather(v)/all-reduce structure the GPU engines use, with the CPU
reference compute step, against the single-process result."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _find_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _dist_pagerank_worker(rank, world, port, scale, ne, seed, iters, outq):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import _native as nat
        from lux_amd import dist as dx
        from lux_amd.graph import Graph

        g = Graph.rmat(scale, ne, seed=seed)
        part = g.partition(world)
        rl, rr, cl, ce, src, _w = g.slice(part, rank)
        vp = part.verts(rank)

        # out-degrees: local histogram + all-reduce (as the GPU engine does)
        deg_t = torch.zeros(g.nv, dtype=torch.int32)
        deg_local = np.bincount(src, minlength=g.nv).astype(np.int32)
        deg_t += torch.from_numpy(deg_local)
        dx.all_reduce_sum_(deg_t)
        deg = deg_t.numpy().view(np.uint32)

        old_t = torch.from_numpy(nat.pagerank_init(g.nv, deg))
        new_t = torch.empty(vp, dtype=torch.float32)
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        for _ in range(iters):
            nat.pagerank_iter_part(g.nv, rl, rr, cl, ce, src, deg,
                                   old_t.numpy(), new_t.numpy())
            dx.all_gather_slices(old_t, new_t, verts, lefts)
        if rank == 0:
            outq.put(old_t.numpy().copy())
    finally:
        dist.destroy_process_group()


def _dist_sssp_worker(rank, world, port, scale, ne, seed, outq):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import _native as nat
        from lux_amd import dist as dx
        from lux_amd.graph import Graph

        g = Graph.rmat(scale, ne, seed=seed)
        part = g.partition(world)
        rl, rr, cl, ce, src, _w = g.slice(part, rank)
        vp = part.verts(rank)
        old_t = torch.full((g.nv,), -1, dtype=torch.int32)
        old_np = old_t.numpy().view(np.uint32)
        old_np[0] = 0
        new_t = torch.empty(vp, dtype=torch.int32)
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        while True:
            changed = nat.sssp_iter_part(rl, rr, cl, ce, src,
                                         old_t.numpy().view(np.uint32),
                                         new_t.numpy().view(np.uint32))
            dx.all_gather_slices(old_t, new_t, verts, lefts)
            ct = torch.tensor([int(changed)])
            dx.all_reduce_sum_(ct)
            if int(ct.item()) == 0:
                break
        if rank == 0:
            outq.put(old_t.numpy().view(np.uint32).copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_dist_pagerank_matches_single(world):
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    scale, ne, seed, iters = 10, 20000, 41, 4
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_pagerank_worker,
             args=(world, _find_port(), scale, ne, seed, iters, outq),
             nprocs=world, join=True)
    got = outq.get()
    g = Graph.rmat(scale, ne, seed=seed)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=1e-6)


@pytest.mark.parametrize("world", [2])
def test_dist_sssp_matches_single(world):
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    scale, ne, seed = 9, 8000, 43
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_sssp_worker,
             args=(world, _find_port(), scale, ne, seed, outq),
             nprocs=world, join=True)
    got = outq.get()
    g = Graph.rmat(scale, ne, seed=seed)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)


def _dist_pagerank_pipelined_worker(rank, world, port, scale, ne, seed,
                                    iters, outq):
    """Publish-early schedule with the async all-gather handle — the
    collective mechanics the pipelined GPU engines use (engine.py
    PagerankEngine.step), with the CPU reference compute."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import _native as nat
        from lux_amd import dist as dx
        from lux_amd.graph import Graph

        g = Graph.rmat(scale, ne, seed=seed)
        part = g.partition(world)
        rl, rr, cl, ce, src, _w = g.slice(part, rank)
        vp = part.verts(rank)
        deg_t = torch.zeros(g.nv, dtype=torch.int32)
        deg_t += torch.from_numpy(
            np.bincount(src, minlength=g.nv).astype(np.int32))
        dx.all_reduce_sum_(deg_t)
        deg = deg_t.numpy().view(np.uint32)
        old_t = torch.from_numpy(nat.pagerank_init(g.nv, deg))
        new_t = torch.empty(vp, dtype=torch.float32)
        cur_t = old_t.narrow(0, int(part.row_left[rank]), vp).clone()
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        handle = None
        for _ in range(iters):
            if handle is not None:
                handle.wait()
            nat.pagerank_iter_part(g.nv, rl, rr, cl, ce, src, deg,
                                   old_t.numpy(), new_t.numpy())
            cur_t.copy_(new_t)
            handle = dx.all_gather_slices_async(old_t, cur_t, verts, lefts)
        handle.wait()
        if rank == 0:
            outq.put(old_t.numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_dist_pagerank_async_pipeline(world):
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    scale, ne, seed, iters = 10, 20000, 47, 4
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_pagerank_pipelined_worker,
             args=(world, _find_port(), scale, ne, seed, iters, outq),
             nprocs=world, join=True)
    got = outq.get()
    g = Graph.rmat(scale, ne, seed=seed)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=1e-6)


def _uf_find(parent, v):
    while parent[v] != v:
        parent[v] = parent[parent[v]]
        v = parent[v]
    return v


def _uf_union(parent, a, b):
    ra, rb = _uf_find(parent, a), _uf_find(parent, b)
    if ra == rb:
        return
    if ra < rb:  # larger id wins: flattened labels = component max
        parent[ra] = rb
    else:
        parent[rb] = ra


def _dist_cc_star_worker(rank, world, port, scale, ne, seed, outq):
    """The CCUnionFindEngine star-forest exchange (cc_engine.py run):
    per-rank union of its edge partition, then all-gather label vectors
    and union the peers' stars until fixpoint — with a CPU union-find."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import dist as dx
        from lux_amd.graph import Graph
        g = Graph.rmat(scale, ne, seed=seed, sym=True)
        part = g.partition(world)
        rl, rr, cl, ce, srcs, _w = g.slice(part, rank)
        parent = np.arange(g.nv, dtype=np.int64)
        counts = np.diff(np.concatenate([[cl], ce])).astype(np.int64)
        dsts = np.repeat(np.arange(rl, rr + 1, dtype=np.int64), counts)
        for s_, d_ in zip(srcs.astype(np.int64), dsts):
            _uf_union(parent, s_, d_)
        labels = np.array([_uf_find(parent, v) for v in range(g.nv)])
        lab_t = torch.from_numpy(labels.astype(np.int32))
        gathered = torch.empty(world * g.nv, dtype=torch.int32)
        rounds = 0
        while True:
            dx.all_gather_slices(gathered, lab_t, [g.nv] * world,
                                 [q * g.nv for q in range(world)])
            gn = gathered.numpy()
            for q in range(world):
                if q != rank:
                    star = gn[q * g.nv:(q + 1) * g.nv]
                    for v in range(g.nv):
                        _uf_union(parent, v, int(star[v]))
            new = np.array([_uf_find(parent, v) for v in range(g.nv)])
            changed = torch.tensor([int((new != labels).any())])
            dx.all_reduce_sum_(changed)
            labels = new
            lab_t = torch.from_numpy(labels.astype(np.int32))
            rounds += 1
            if int(changed.item()) == 0:
                break
        if rank == 0:
            outq.put((labels.copy(), rounds))
    finally:
        dist.destroy_process_group()


def test_dist_cc_star_forest(world=4):
    from lux_amd.graph import Graph
    scale, ne, seed = 9, 4000, 23
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_cc_star_worker,
             args=(world, _find_port(), scale, ne, seed, outq),
             nprocs=world, join=True)
    got, rounds = outq.get()
    # expected: component max over the SAME undirected graph
    g = Graph.rmat(scale, ne, seed=seed, sym=True)
    parent = np.arange(g.nv, dtype=np.int64)
    counts = np.diff(np.concatenate([[0], g.col_end])).astype(np.int64)
    dsts = np.repeat(np.arange(g.nv, dtype=np.int64), counts)
    for s_, d_ in zip(g.src.astype(np.int64), dsts):
        _uf_union(parent, s_, d_)
    want = np.array([_uf_find(parent, v) for v in range(g.nv)])
    assert np.array_equal(got, want)
    assert rounds <= 4  # monotone merges: O(log P) rounds


def _dist_cf_slices_worker(rank, world, port, outq):
    """CF latent-vector exchange shape (cf_engine.py step): uneven
    K-scaled element slices of the [nv, K] factor matrix."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import dist as dx
        nv, K = 10, 4
        verts = [3, 7]
        lefts = [0, 3]
        old = torch.arange(nv * K, dtype=torch.float32)
        my = old.narrow(0, lefts[rank] * K, verts[rank] * K).clone()
        my *= (rank + 1) * 10.0
        for _ in range(3):  # repeated sweeps reuse the same buffers
            dx.all_gather_slices(old, my, [v * K for v in verts],
                                 [off * K for off in lefts])
        if rank == 0:
            outq.put(old.numpy().copy())
    finally:
        dist.destroy_process_group()


def test_dist_cf_vector_slices(world=2):
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_cf_slices_worker, args=(world, _find_port(), outq),
             nprocs=world, join=True)
    got = outq.get()
    want = np.arange(40, dtype=np.float32)
    want[:12] *= 10.0
    want[12:] *= 20.0
    np.testing.assert_array_equal(got, want)


def _dist_multi_worker(rank, world, port, outq):
    """exchange_multi_async: one batched p2p round carrying a byte-segment
    part AND a label part where rank 1 skips publishing (the push engine's
    sparse-iteration label skip)."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import dist as dx
        seg_used = [24, 40]
        seg_off = [0, 48]
        segs = torch.zeros(96, dtype=torch.uint8)
        my_seg = torch.full((seg_used[rank],), 50 + rank, dtype=torch.uint8)
        lab_n = [5, 0]  # rank 1 skips its label publish
        lab_off = [0, 5]
        labels = torch.full((9,), -1, dtype=torch.int32)
        my_lab = torch.full((5,), 100 + rank, dtype=torch.int32) \
            if rank == 0 else None
        h = dx.exchange_multi_async([
            (segs, my_seg, seg_used, seg_off),
            (labels, my_lab, lab_n, lab_off)])
        h.wait()
        if rank == 0:
            outq.put((segs.numpy().copy(), labels.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_dist_exchange_multi(world=2):
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_multi_worker, args=(world, _find_port(), outq),
             nprocs=world, join=True)
    segs, labels = outq.get()
    assert (segs[:24] == 50).all() and (segs[24:48] == 0).all()
    assert (segs[48:88] == 51).all() and (segs[88:] == 0).all()
    assert (labels[:5] == 100).all()
    assert (labels[5:] == -1).all()  # skipped publisher left stale values


def _dist_bytes_worker(rank, world, port, outq):
    """Uneven uint8 segment exchange — the frontier (header+payload byte
    segment) all-gather shape PushEngine uses (push_engine.py step)."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import dist as dx
        seg_bytes = [48, 96]  # uneven, 16-aligned like _align16 segments
        off = [0, 48]
        total = sum(seg_bytes)
        mine = torch.full((seg_bytes[rank],), 10 + rank, dtype=torch.uint8)
        mine[0] = 7  # "header" byte
        full = torch.zeros(total, dtype=torch.uint8)
        dx.all_gather_slices(full, mine, seg_bytes, off)
        if rank == 0:
            outq.put(full.numpy().copy())
    finally:
        dist.destroy_process_group()


def test_dist_byte_segments(world=2):
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_bytes_worker, args=(world, _find_port(), outq),
             nprocs=world, join=True)
    got = outq.get()
    assert got[0] == 7 and got[48] == 7
    assert (got[1:48] == 10).all() and (got[49:] == 11).all()


def _dist_push_payload_worker(rank, world, port, outq):
    """The push engine's per-iteration payload exchange
    (push_engine.exchange_frontier_payloads) with synthetic meta/segments:
    rank 0 dense (publishes bitmap bytes + label slice), rank 1 sparse
    (publishes used queue bytes + label annex, label slice SKIPPED)."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        import torch

        from lux_amd.push_engine import exchange_frontier_payloads
        from lux_amd.types import DENSE_BITMAP, SPARSE_QUEUE
        verts_all = [20, 12]
        row_left_all = [0, 20]
        seg_bytes = [64, 64]
        seg_off = [0, 64]
        annex_off = [0, 16]
        meta = np.zeros((2, 8), np.uint32)
        meta[0, 0], meta[0, 1] = DENSE_BITMAP, 5
        meta[1, 0], meta[1, 1] = SPARSE_QUEUE, 3
        fq_all = torch.zeros(128, dtype=torch.uint8)
        annex_all = torch.zeros(32, dtype=torch.int32)
        labels = torch.full((32,), -1, dtype=torch.int32)
        labels_part = torch.full((verts_all[rank],), 70 + rank,
                                 dtype=torch.int32)
        if rank == 0:  # dense seg: 8B hdr + ceil(20/8)=3 bitmap bytes
            new_seg = torch.full((64,), 0xA0, dtype=torch.uint8)
            new_seg[:8].view(torch.int32)[0] = DENSE_BITMAP
            new_seg[:8].view(torch.int32)[1] = 5
            new_annex = torch.zeros(16, dtype=torch.int32)
        else:  # sparse: 8B hdr + 3*4 queue ids; annex carries 3 labels
            new_seg = torch.full((64,), 0xB1, dtype=torch.uint8)
            new_seg[:8].view(torch.int32)[0] = SPARSE_QUEUE
            new_seg[:8].view(torch.int32)[1] = 3
            new_annex = torch.arange(100, 116, dtype=torch.int32)
        h, lab_n = exchange_frontier_payloads(
            meta, verts_all, fq_all, new_seg, seg_off, annex_all, new_annex,
            annex_off, labels, labels_part, row_left_all, rank)
        h.wait()
        assert lab_n == [20, 0]
        if rank == 0:
            outq.put((fq_all.numpy().copy(), annex_all.numpy().copy(),
                      labels.numpy().copy()))
    finally:
        dist.destroy_process_group()


def test_dist_push_payload_exchange(world=2):
    from lux_amd.types import DENSE_BITMAP, SPARSE_QUEUE
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_push_payload_worker, args=(world, _find_port(), outq),
             nprocs=world, join=True)
    fq, annex, labels = outq.get()
    # rank 0's dense seg: used = 8 + 3 = 11 bytes of 0xA0 payload
    assert fq[:4].view(np.uint32)[0] == DENSE_BITMAP
    assert (fq[8:11] == 0xA0).all()
    assert (fq[11:64] == 0).all()  # beyond used bytes: never shipped
    # rank 1's sparse seg at offset 64: used = 8 + 12
    assert fq[64:68].view(np.uint32)[0] == SPARSE_QUEUE
    assert (fq[72:84] == 0xB1).all()
    assert (fq[84:] == 0).all()
    # annex: only rank 1's 3 labels (at its annex offset 16)
    assert (annex[:16] == 0).all()
    assert list(annex[16:19]) == [100, 101, 102]
    assert (annex[19:] == 0).all()
    # labels: rank 0 (dense) published its slice; rank 1 (sparse) did not
    assert (labels[:20] == 70).all()
    assert (labels[20:] == -1).all()


def _dist_halo_pagerank_worker(rank, world, port, iters, outq):
    """PageRank over a ring-of-cliques graph exchanged through the
    in_vtxs halo machinery (halo.py, LUX_HALO=1): only boundary vertices
    travel. Verifies the packed/scattered values reproduce the full
    all-gather result exactly."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["LUX_HALO"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import _native as nat
        from lux_amd import dist as dx
        from lux_amd.graph import Graph
        from lux_amd.halo import HaloExchange
        # ring graph + a few local chords: halo per partition ~ 1 vertex
        nv = 4096
        src = list(range(nv)) + [v for v in range(0, nv, 4)]
        dst = [(v + 1) % nv for v in range(nv)] + \
            [(v + 2) % nv for v in range(0, nv, 4)]
        g = Graph.from_edges(nv, src, dst)
        part = g.partition(world)
        rl, rr, cl, ce, srcs, _w = g.slice(part, rank)
        vp = part.verts(rank)
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        col_t = torch.from_numpy(srcs.astype(np.int32))
        halo = HaloExchange(g.nv, lefts, verts, rank, col_t)
        assert halo.worth_it()
        assert halo.total_halo() < nv // world  # boundary-sized, not nv
        deg_t = torch.zeros(g.nv, dtype=torch.int32)
        deg_t += torch.from_numpy(
            np.bincount(srcs, minlength=g.nv).astype(np.int32))
        dx.all_reduce_sum_(deg_t)
        deg = deg_t.numpy().view(np.uint32)
        old_t = torch.from_numpy(nat.pagerank_init(g.nv, deg))
        new_t = torch.empty(vp, dtype=torch.float32)
        for _ in range(iters):
            nat.pagerank_iter_part(g.nv, rl, rr, cl, ce, srcs, deg,
                                   old_t.numpy(), new_t.numpy())
            halo.publish(old_t, new_t)
        # finalize: one full gather so every position is current
        dx.all_gather_slices(old_t, old_t.narrow(0, lefts[rank], vp),
                             verts, lefts)
        if rank == 0:
            outq.put(old_t.numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
def test_dist_halo_pagerank(world):
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    iters = 5
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_halo_pagerank_worker,
             args=(world, _find_port(), iters, outq), nprocs=world,
             join=True)
    got = outq.get()
    nv = 4096
    src = list(range(nv)) + [v for v in range(0, nv, 4)]
    dst = [(v + 1) % nv for v in range(nv)] + \
        [(v + 2) % nv for v in range(0, nv, 4)]
    g = Graph.from_edges(nv, src, dst)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=1e-6)


def _dist_halo_autodisable_worker(rank, world, port, outq):
    """On an RMAT graph the halo covers most of nv — worth_it() must say
    no (the engines then keep the plain slice all-gather)."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.pop("LUX_HALO", None)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd.graph import Graph
        from lux_amd.halo import HaloExchange
        g = Graph.rmat(10, 30000, seed=3)
        part = g.partition(world)
        _rl, _rr, _cl, _ce, srcs, _w = g.slice(part, rank)
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        halo = HaloExchange(g.nv, lefts, verts, rank,
                            torch.from_numpy(srcs.astype(np.int32)))
        if rank == 0:
            outq.put(bool(halo.worth_it()))
    finally:
        dist.destroy_process_group()


def test_dist_halo_autodisable_on_rmat(world=2):
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_halo_autodisable_worker, args=(world, _find_port(), outq),
             nprocs=world, join=True)
    assert outq.get() is False


def _dist_per_peer_worker(rank, world, port, scale, ne, seed, iters, outq):
    """all_gather_slices_per_peer: one p2p group per peer, consumed in
    ring-offset order (the pipelined pull engines' per-peer remote
    sweeps). Equivalence vs the single-process PageRank reference."""
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from lux_amd import _native as nat
        from lux_amd import dist as dx
        from lux_amd.graph import Graph
        g = Graph.rmat(scale, ne, seed=seed)
        part = g.partition(world)
        rl, rr, cl, ce, src, _w = g.slice(part, rank)
        vp = part.verts(rank)
        deg_t = torch.zeros(g.nv, dtype=torch.int32)
        deg_t += torch.from_numpy(
            np.bincount(src, minlength=g.nv).astype(np.int32))
        dx.all_reduce_sum_(deg_t)
        deg = deg_t.numpy().view(np.uint32)
        old_t = torch.from_numpy(nat.pagerank_init(g.nv, deg))
        new_t = torch.empty(vp, dtype=torch.float32)
        verts = [part.verts(p) for p in range(world)]
        lefts = [int(part.row_left[p]) for p in range(world)]
        handles = []
        for _ in range(iters):
            # wait each peer's group before "using" its slice
            for _q, w in handles:
                w.wait()
            nat.pagerank_iter_part(g.nv, rl, rr, cl, ce, src, deg,
                                   old_t.numpy(), new_t.numpy())
            old_t.narrow(0, lefts[rank], vp).copy_(new_t)
            handles = dx.all_gather_slices_per_peer(
                old_t, old_t.narrow(0, lefts[rank], vp), verts, lefts)
        for _q, w in handles:
            w.wait()
        if rank == 0:
            outq.put(old_t.numpy().copy())
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 3, 4])
def test_dist_per_peer_exchange(world):
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    scale, ne, seed, iters = 10, 20000, 53, 4
    ctx = mp.get_context("spawn")
    outq = ctx.SimpleQueue()
    mp.spawn(_dist_per_peer_worker,
             args=(world, _find_port(), scale, ne, seed, iters, outq),
             nprocs=world, join=True)
    got = outq.get()
    g = Graph.rmat(scale, ne, seed=seed)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=1e-6)
