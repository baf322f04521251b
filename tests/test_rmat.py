"""R-MAT generator tests: determinism, shape, skew."""
import numpy as np

from lux_amd import _native as nat
from lux_amd.graph import Graph


def test_deterministic():
    s1, d1 = nat.rmat_edges(42, 10, 5000)
    s2, d2 = nat.rmat_edges(42, 10, 5000)
    assert np.array_equal(s1, s2) and np.array_equal(d1, d2)
    s3, _ = nat.rmat_edges(43, 10, 5000)
    assert not np.array_equal(s1, s3)


def test_ids_in_range():
    s, d = nat.rmat_edges(1, 8, 10000)
    assert s.max() < 256 and d.max() < 256


def test_skew():
    """RMAT(a=0.57) must be heavy-tailed: the top 1% of vertices should own
    far more than 1% of the edges."""
    g = Graph.rmat(12, 1 << 16, seed=5)
    deg = np.diff(np.concatenate([[0], g.col_end])).astype(np.int64)
    top = np.sort(deg)[::-1][: g.nv // 100].sum()
    assert top > 0.15 * g.ne


def test_csc_consistency():
    g = Graph.rmat(8, 3000, seed=9)
    assert int(g.col_end[-1]) == g.ne
    assert np.all(np.diff(g.col_end.astype(np.int64)) >= 0)
    # col_end vs explicit histogram of dsts
    s, d = nat.rmat_edges(9, 8, 3000)
    hist = np.bincount(d, minlength=g.nv).cumsum()
    assert np.array_equal(g.col_end, hist.astype(np.uint64))


def test_bipartite_weighted():
    g = Graph.bipartite(1000, 256, 20000, seed=3)
    assert g.nv == 1256
    assert g.weight is not None
    assert g.weight.min() >= 1 and g.weight.max() <= 5
    # each rating is stored in BOTH directions (reference NetFlix parity:
    # the .nf edge file holds user->item and item->user copies), so edges
    # split evenly and both sides have in-edges
    assert g.row_end(999) == g.ne // 2      # user in-edges = item->user half
    assert g.row_end(1255) == g.ne          # item in-edges = user->item half


def test_bipartite_pair_symmetry():
    """Even/odd generator indices form one rating: reversed endpoints,
    identical weight (rmat.h bipartite_edge pair emission)."""
    src, dst, w = nat.bipartite_edges(7, 500, 128, 10000)
    assert np.array_equal(src[0::2], dst[1::2])
    assert np.array_equal(dst[0::2], src[1::2])
    assert np.array_equal(w[0::2], w[1::2])
    # direction check: even edges go user->item, odd item->user
    assert src[0::2].max() < 500 and dst[0::2].min() >= 500


def test_bipartite_pairs_fuzz():
    """Pair invariants over random shapes: endpoints reversed, shared
    weight, ids in range, and the dst histogram splits exactly ne/2 per
    side (every rating contributes one in-edge to each side)."""
    rng = np.random.default_rng(11)
    for _ in range(5):
        nu = int(rng.integers(10, 3000))
        ni = int(rng.integers(2, 500))
        ne = int(rng.integers(2, 20000)) * 2
        seed = int(rng.integers(1, 1 << 30))
        src, dst, w = nat.bipartite_edges(seed, nu, ni, ne)
        assert np.array_equal(src[0::2], dst[1::2])
        assert np.array_equal(dst[0::2], src[1::2])
        assert np.array_equal(w[0::2], w[1::2])
        assert src[0::2].max() < nu and dst[1::2].max() < nu
        assert dst[0::2].min() >= nu and dst[0::2].max() < nu + ni
        assert (dst < nu).sum() == ne // 2  # item->user half
