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
