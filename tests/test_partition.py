"""Edge-balanced partitioner tests (reference rule: core/pull_model.inl:108-131)."""
import numpy as np
import pytest

from lux_amd.graph import Graph


@pytest.mark.parametrize("nparts", [1, 2, 3, 8])
def test_disjoint_complete(nparts):
    g = Graph.rmat(10, 10000, seed=7)
    part = g.partition(nparts)
    seen = np.zeros(g.nv, bool)
    for p in range(nparts):
        if part.verts(p) == 0:
            continue
        rl, rr = int(part.row_left[p]), int(part.row_right[p])
        assert not seen[rl:rr + 1].any(), "overlapping partitions"
        seen[rl:rr + 1] = True
    assert seen.all(), "vertices uncovered"


@pytest.mark.parametrize("nparts", [2, 4, 8])
def test_edge_balance(nparts):
    g = Graph.rmat(12, 50000, seed=11)
    part = g.partition(nparts)
    cap = (g.ne + nparts - 1) // nparts
    max_deg = int(np.max(np.diff(np.concatenate([[0], g.col_end]))))
    for p in range(nparts):
        # each part carries at most cap + one vertex's worth of slack
        assert part.edges(p) <= cap + max_deg


def test_contiguous_edge_ranges():
    g = Graph.rmat(9, 4000, seed=2)
    part = g.partition(4)
    prev = 0
    for p in range(4):
        if part.verts(p) == 0:
            assert part.edges(p) == 0
            continue
        assert int(part.col_left[p]) == prev
        prev = int(part.col_right[p])
    assert prev == g.ne


def test_more_parts_than_vertices():
    g = Graph.from_edges(3, [0, 1], [1, 2])
    part = g.partition(8)
    total_v = sum(part.verts(p) for p in range(8))
    total_e = sum(part.edges(p) for p in range(8))
    assert total_v == 3
    assert total_e == 2


def test_hub_vertex_gets_own_partition():
    # one vertex with 1000 in-edges among tiny ones
    src = list(range(1, 1001)) + [0, 0]
    dst = [0] * 1000 + [1, 2]
    g = Graph.from_edges(1024, src, dst)
    part = g.partition(4)
    # partition containing vertex 0 must still terminate & cover everything
    total = sum(part.verts(p) for p in range(4))
    assert total == g.nv
