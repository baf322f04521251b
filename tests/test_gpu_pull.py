"""GPU tests: graph builder equivalence vs CPU, device scan, pull engine
numerics vs the CPU fp32 reference. All marked gpu."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from lux_amd import _native as nat  # noqa: E402
from lux_amd import _native_gpu as ng  # noqa: E402
from lux_amd import cpu_ref  # noqa: E402
from lux_amd.engine import (DeviceCSC, GraphPart, LabelPullEngine,  # noqa
                            PagerankEngine, partition_bounds)
from lux_amd.graph import Graph  # noqa: E402

U32, U64, F32 = torch.int32, torch.int64, torch.float32


def stream():
    return torch.cuda.current_stream().cuda_stream


def test_native_lib_is_loaded():
    """The HIP extension must be the in-tree .so (no silent fallback)."""
    path = ng.lib()._name
    assert path.endswith("liblux_gpu.so")


def test_rmat_edges_match_cpu():
    ne, scale, seed = 100000, 14, 9
    cs, cd = nat.rmat_edges(seed, scale, ne)
    gs = torch.empty(ne, dtype=U32, device="cuda")
    gd = torch.empty(ne, dtype=U32, device="cuda")
    ng.rmat_edges(stream(), seed, scale, ne, gs, gd)
    torch.cuda.synchronize()
    assert np.array_equal(gs.cpu().numpy().view(np.uint32), cs)
    assert np.array_equal(gd.cpu().numpy().view(np.uint32), cd)


def test_device_scan():
    n = 1 << 20
    rng = np.random.default_rng(3)
    vals = rng.integers(0, 1000, n, dtype=np.uint32)
    inp = torch.from_numpy(vals.view(np.int32)).cuda()
    out = torch.empty(n, dtype=U64, device="cuda")
    partials = torch.empty(ng.scan_partials_size(n), dtype=U64, device="cuda")
    ng.scan_end_offsets(stream(), n, inp, out, partials)
    torch.cuda.synchronize()
    want = vals.astype(np.uint64).cumsum()
    assert np.array_equal(out.cpu().numpy().view(np.uint64), want)


def test_gpu_csc_matches_cpu():
    scale, ne = 12, 200000
    gg = DeviceCSC.rmat(scale, ne, seed=5)
    torch.cuda.synchronize()
    g = Graph.rmat(scale, ne, seed=5)
    assert np.array_equal(gg.col_end.cpu().numpy().view(np.uint64), g.col_end)
    # per-dst source multisets equal (scatter order differs)
    gsrc = gg.src.cpu().numpy().view(np.uint32)
    b = 0
    for v in range(g.nv):
        e = int(g.col_end[v])
        assert sorted(gsrc[b:e]) == sorted(g.src[b:e]), f"vertex {v}"
        b = e


def test_partition_bounds_match_cpu():
    scale, ne = 13, 300000
    gg = DeviceCSC.rmat(scale, ne, seed=8)
    g = Graph.rmat(scale, ne, seed=8)
    for P in (1, 2, 4, 8):
        rl, rr = partition_bounds(gg.col_end, ne, P)
        part = g.partition(P)
        for p in range(P):
            if part.verts(p) == 0:
                assert rr[p] < rl[p]
            else:
                assert rl[p] == part.row_left[p]
                assert rr[p] == part.row_right[p]


@pytest.mark.parametrize("scale,ne", [(10, 10000), (14, 1 << 20)])
def test_pagerank_single_part_vs_cpu(scale, ne):
    iters = 5
    full = DeviceCSC.rmat(scale, ne, seed=11)
    part = GraphPart(full, 1, 0)
    eng = PagerankEngine(part)
    for _ in range(iters):
        eng.step()
    got = eng.ranks().cpu().numpy()
    g = Graph.rmat(scale, ne, seed=11)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-9)


def test_pagerank_multipart_single_process():
    """4 partitions stepped in one process — validates the partitioned
    kernels compose to the whole-graph result (the N-GPU equivalence test
    shape from SURVEY.md §4(d), runnable on one GPU)."""
    scale, ne, iters, P = 12, 300000, 4, 4
    full = DeviceCSC.rmat(scale, ne, seed=13)
    parts = [GraphPart(full, P, p, keep_full=True) for p in range(P)]
    for pt in parts:
        pt.build_bins()
    nv = 1 << scale
    deg = torch.zeros(nv, dtype=U32, device="cuda")
    for pt in parts:
        ng.hist_u32(stream(), pt.ep, pt.col, deg)
    rank0 = 1.0 / nv
    degf = deg.to(F32)
    old = torch.where(deg == 0, torch.full_like(degf, rank0),
                      rank0 / degf.clamp(min=1.0))
    new = torch.empty_like(old)
    init_rank = (1 - 0.15) / nv
    from lux_amd.engine import run_pull
    for _ in range(iters):
        for pt in parts:
            if pt.vp == 0:
                continue
            run_pull(pt, ng.PULL_PR, old,
                     new.narrow(0, pt.row_left, pt.vp), deg, init_rank)
        old, new = new, old
    g = Graph.rmat(scale, ne, seed=13)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(old.cpu().numpy(), want, rtol=2e-4, atol=1e-9)


def test_blocked_pagerank_matches_unblocked():
    """src-blocked CSC (forced 16 blocks on a small graph) must reproduce
    the plain single-sweep results bit-for-bit-ish (same fp32 order within
    rows differs; compare with tolerance)."""
    scale, ne, iters = 12, 300000, 4
    full = DeviceCSC.rmat(scale, ne, seed=31)
    part = GraphPart(full, 1, 0)
    part.prepare_pull(force_shift=scale - 4)  # 16 src blocks
    assert part.blocks is not None and len(part.blocks) == 16
    eng = PagerankEngine(part)
    for _ in range(iters):
        eng.step()
    got = eng.ranks().cpu().numpy()
    g = Graph.rmat(scale, ne, seed=31)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-9)


def test_blocked_label_pull_matches():
    scale, ne = 11, 60000
    full = DeviceCSC.rmat(scale, ne, seed=33)
    part = GraphPart(full, 1, 0)
    part.prepare_pull(force_shift=scale - 3)  # 8 blocks
    assert part.blocks
    nv = 1 << scale
    init = torch.full((nv,), -1, dtype=U32, device="cuda")
    init[0] = 0
    eng = LabelPullEngine(part, ng.PULL_MIN, init)
    eng.run_to_fixpoint(max_iters=nv)
    got = eng.old.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=33)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)


def test_label_pull_sssp_dense_vs_cpu():
    scale, ne = 11, 60000
    full = DeviceCSC.rmat(scale, ne, seed=17)
    part = GraphPart(full, 1, 0)
    nv = 1 << scale
    init = torch.full((nv,), -1, dtype=U32, device="cuda")  # 0xFFFFFFFF
    init[0] = 0
    eng = LabelPullEngine(part, ng.PULL_MIN, init)
    eng.run_to_fixpoint(max_iters=nv)
    got = eng.old.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=17)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)
    assert cpu_ref.sssp_check(g, got) == 0


def test_label_pull_cc_vs_cpu():
    scale, ne = 11, 60000
    full = DeviceCSC.rmat(scale, ne, seed=19)
    part = GraphPart(full, 1, 0)
    nv = 1 << scale
    init = torch.arange(nv, dtype=U32, device="cuda")
    eng = LabelPullEngine(part, ng.PULL_MAX, init)
    eng.run_to_fixpoint(max_iters=nv)
    got = eng.old.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=19)
    want, _ = cpu_ref.cc(g)
    assert np.array_equal(got, want)


def test_pipelined_subset_schedule_matches_cpu():
    """Simulates the pipelined multi-rank PR schedule in one process:
    rank-aligned blocked CSC, local block swept against the rank's own
    slice via the offset base pointer, remote blocks against the gathered
    array — the exact kernel sequence PagerankEngine runs per rank when
    world_size > 1 (with the RCCL all-gather replaced by device copies)."""
    from lux_amd.engine import run_pull_sweeps
    scale, ne, iters, P = 12, 300000, 3, 2
    nv = 1 << scale
    full = DeviceCSC.rmat(scale, ne, seed=17)
    parts = [GraphPart(full, P, p, keep_full=(p < P - 1)) for p in range(P)]
    for pt in parts:
        pt.prepare_pull(force_shift=scale - 3)  # 8 windows + rank bounds
        assert pt.blocks is not None
        assert any(b["local"] for b in pt.blocks)
        assert any(not b["local"] for b in pt.blocks)
    deg = torch.zeros(nv, dtype=U32, device="cuda")
    for pt in parts:
        ng.hist_u32(stream(), pt.ep, pt.col, deg)
    rank0 = 1.0 / nv
    degf = deg.to(F32)
    old = torch.where(deg == 0, torch.full_like(degf, rank0),
                      rank0 / degf.clamp(min=1.0))
    init_rank = (1 - 0.15) / nv
    curs = [old.narrow(0, pt.row_left, pt.vp).clone() for pt in parts]
    news = [torch.empty(pt.vp, dtype=F32, device="cuda") for pt in parts]
    for _ in range(iters):
        for pt, cur, new in zip(parts, curs, news):
            new.zero_()
            base = cur.data_ptr() - pt.row_left * 4
            run_pull_sweeps(pt, ng.PULL_PR, base, new, deg, init_rank,
                            subset="local")
            run_pull_sweeps(pt, ng.PULL_PR, old, new, deg, init_rank,
                            subset="remote")
            ng.pull_finish_pr(stream(), pt.vp, new, deg, pt.row_left,
                              init_rank)
        for pt, new, cur in zip(parts, news, curs):
            cur.copy_(new)
            old.narrow(0, pt.row_left, pt.vp).copy_(new)
    g = Graph.rmat(scale, ne, seed=17)
    want = cpu_ref.pagerank(g, iters)
    np.testing.assert_allclose(old.cpu().numpy(), want, rtol=2e-4,
                               atol=1e-9)


def test_empty_graph_engines():
    """0-edge graph: engines must not crash; PR gives the uniform rank,
    labels stay at init."""
    nv = 1024
    col_end = torch.zeros(nv, dtype=torch.int64, device="cuda")
    src = torch.zeros(1, dtype=U32, device="cuda")
    full = DeviceCSC(nv, 0, col_end, src)
    part = GraphPart(full, 1, 0)
    eng = PagerankEngine(part)
    eng.step()
    r = eng.ranks().cpu().numpy()
    # zero in-sum everywhere: pr = (1-alpha)/nv (stored undivided, deg=0)
    np.testing.assert_allclose(r, np.full(nv, 0.85 / nv, np.float32),
                               rtol=1e-6)


def test_self_loops_and_duplicates():
    """Self-loops and duplicate edges flow through build + engines (the
    reference neither filters nor documents them; we keep them as-is)."""
    from lux_amd.graph import Graph
    src = [0, 0, 1, 2, 2, 2, 3]
    dst = [0, 1, 1, 2, 2, 0, 3]
    g = Graph.from_edges(8, src, dst)
    want = cpu_ref.pagerank(g, 3)
    s = torch.tensor(src, dtype=U32, device="cuda")
    d = torch.tensor(dst, dtype=U32, device="cuda")
    full = DeviceCSC._from_device_edges(8, 7, s, d, None, "cuda")
    eng = PagerankEngine(GraphPart(full, 1, 0))
    for _ in range(3):
        eng.step()
    np.testing.assert_allclose(eng.ranks().cpu().numpy(), want, rtol=2e-4,
                               atol=1e-9)


def test_blocked_build_grouped_matches(monkeypatch):
    """Grouped blocked build (tiny slot budget forces one window per
    group) must produce the same PageRank results as the single-pass
    build."""
    from lux_amd.engine import DeviceCSC, GraphPart, PagerankEngine
    scale, ne, seed, iters = 12, 150000, 7, 4
    monkeypatch.setenv("LUX_BLOCK_SHIFT", "9")  # force many windows
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    eng = PagerankEngine(GraphPart(full, 1, 0, keep_full=True))
    assert eng.part.blocks and len(eng.part.blocks) > 4
    for _ in range(iters):
        eng.step()
    want = eng.ranks().cpu().numpy().copy()
    monkeypatch.setenv("LUX_BLOCK_GROUP_SLOTS", str(1 << scale))  # 1/group
    eng2 = PagerankEngine(GraphPart(full, 1, 0))
    assert len(eng2.part.blocks) == len(eng.part.blocks)
    for _ in range(iters):
        eng2.step()
    got = eng2.ranks().cpu().numpy()
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-12)
