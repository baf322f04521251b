"""Tests for the .lux binary format (parity with reference README.md:56-75)."""
import struct

import numpy as np

from lux_amd.graph import Graph


def small_graph():
    # 0->1, 0->2, 1->2, 2->0, 3->2 (src, dst)
    return Graph.from_edges(4, [0, 0, 1, 2, 3], [1, 2, 2, 0, 2])


def test_roundtrip(tmp_path):
    g = small_graph()
    p = str(tmp_path / "g.lux")
    g.save(p)
    g2 = Graph.load(p)
    assert g2.nv == g.nv and g2.ne == g.ne
    assert np.array_equal(g2.col_end, g.col_end)
    assert np.array_equal(g2.src, g.src)


def test_byte_layout(tmp_path):
    """The on-disk bytes are exactly u32 nv, u64 ne, u64 col_end[nv],
    u32 src[ne] — the reference loader's expectation
    (core/pull_model.inl:97-103, :253-320)."""
    g = small_graph()
    p = str(tmp_path / "g.lux")
    g.save(p)
    raw = open(p, "rb").read()
    assert len(raw) == 4 + 8 + 8 * g.nv + 4 * g.ne
    nv, ne = struct.unpack_from("<IQ", raw, 0)
    assert nv == 4 and ne == 5
    col_end = np.frombuffer(raw, np.uint64, nv, 12)
    src = np.frombuffer(raw, np.uint32, ne, 12 + 8 * nv)
    assert np.array_equal(col_end, g.col_end)
    assert np.array_equal(src, g.src)


def test_csc_grouping():
    g = small_graph()
    # dst 0 has in-edge from 2; dst 1 from 0; dst 2 from {0,1,3}; dst 3 none
    assert g.row_begin(0) == 0 and g.row_end(0) == 1
    assert g.src[0] == 2
    assert g.row_end(1) == 2 and g.src[1] == 0
    assert sorted(g.src[g.row_begin(2):g.row_end(2)]) == [0, 1, 3]
    assert g.row_begin(3) == g.row_end(3) == 5


def test_weighted_roundtrip(tmp_path):
    g = Graph.from_edges(3, [0, 1, 2], [1, 2, 0], weight=[7, -3, 11])
    p = str(tmp_path / "w.lux")
    g.save(p)
    g2 = Graph.load(p, want_weights=True)
    assert g2.weight is not None
    # weights permuted identically to sources by the dst counting sort
    for v in range(3):
        b, e = g2.row_begin(v), g2.row_end(v)
        pairs = sorted(zip(g2.src[b:e], g2.weight[b:e]))
        b0, e0 = g.row_begin(v), g.row_end(v)
        assert pairs == sorted(zip(g.src[b0:e0], g.weight[b0:e0]))


def test_tolerates_trailing_degrees(tmp_path):
    """Reference converter appends an unread u32 degree[nv] block
    (tools/converter.cc:108-124); our reader must accept such files."""
    g = small_graph()
    p = str(tmp_path / "t.lux")
    g.save(p)
    with open(p, "ab") as f:
        f.write(np.zeros(g.nv, np.uint32).tobytes())
    g2 = Graph.load(p)
    assert np.array_equal(g2.src, g.src)


def test_partition_slices_cover_graph():
    g = Graph.rmat(8, 2000, seed=3)
    part = g.partition(3)
    covered_v = 0
    covered_e = 0
    for i in range(3):
        if part.verts(i) == 0:
            continue
        rl, rr, cl, ce, src, _w = g.slice(part, i)
        assert cl == g.row_begin(rl)
        assert int(ce[-1]) == g.row_end(rr)
        assert len(src) == part.edges(i)
        covered_v += part.verts(i)
        covered_e += part.edges(i)
    assert covered_v == g.nv
    assert covered_e == g.ne


def test_load_sliced_matches_full_slices(tmp_path):
    """GraphPart.load_sliced (per-partition fseeko read) == slicing the
    fully-loaded graph, for every partition, weighted and not."""
    import numpy as np

    from lux_amd.engine import GraphPart
    from lux_amd.graph import Graph
    g = Graph.rmat(9, 4000, seed=31)
    path = str(tmp_path / "g.lux")
    g.save(path)
    P = 3
    part = g.partition(P)
    for p in range(P):
        gp = GraphPart.load_sliced(path, P, p, device="cpu")
        rl, rr, cl, ce, srcs, _w = g.slice(part, p)
        assert gp.row_left == rl and gp.row_right == rr
        assert gp.col_left == cl
        assert gp.ep == len(srcs)
        rp = gp.row_ptr.numpy().view(np.uint64)
        assert rp[0] == 0 and rp[-1] == gp.ep
        np.testing.assert_array_equal(rp[1:], ce - np.uint64(cl))
        np.testing.assert_array_equal(gp.col.numpy().view(np.uint32), srcs)

    gw = Graph.bipartite(200, 32, 3000, seed=7)
    wpath = str(tmp_path / "w.lux")
    gw.save(wpath)
    partw = gw.partition(2)
    for p in range(2):
        gp = GraphPart.load_sliced(wpath, 2, p, device="cpu",
                                   want_weights=True)
        rl, rr, cl, ce, srcs, w = gw.slice(partw, p)
        np.testing.assert_array_equal(gp.col.numpy().view(np.uint32), srcs)
        np.testing.assert_array_equal(gp.weight.numpy(), w)
