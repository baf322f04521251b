"""Property-based GPU-vs-CPU equivalence on random small graphs: exercises
bin boundaries (deg 0 / T1 / T2 edges), ragged tiles, empty frontiers and
arbitrary sources that fixed cases miss."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

from hypothesis import given, settings, strategies as st  # noqa: E402

from lux_amd import cpu_ref  # noqa: E402
from lux_amd.engine import DeviceCSC, GraphPart, PagerankEngine  # noqa: E402
from lux_amd.graph import Graph  # noqa: E402


@given(scale=st.integers(6, 12), ne=st.integers(0, 20000),
       seed=st.integers(0, 100))
@settings(max_examples=12, deadline=None)
def test_pagerank_fuzz(scale, ne, seed):
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    eng = PagerankEngine(GraphPart(full, 1, 0))
    for _ in range(3):
        eng.step()
    got = eng.ranks().cpu().numpy()
    want = cpu_ref.pagerank(Graph.rmat(scale, ne, seed=seed), 3)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-9)


@given(scale=st.integers(6, 12), ne=st.integers(1, 15000),
       seed=st.integers(0, 100), src_pick=st.integers(0, 1 << 20))
@settings(max_examples=12, deadline=None)
def test_sssp_fuzz(scale, ne, seed, src_pick):
    from lux_amd.push_engine import PushEngine
    nv = 1 << scale
    source = src_pick % nv
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    eng = PushEngine(GraphPart(full, 1, 0), PushEngine.MODE_MIN,
                     source=source)
    eng.run(max_iters=nv)
    got = eng.labels.cpu().numpy().view(np.uint32)
    want, _ = cpu_ref.sssp(Graph.rmat(scale, ne, seed=seed), source)
    np.testing.assert_array_equal(got, want)
    assert eng.check() == 0


@given(scale=st.integers(6, 12), ne=st.integers(2, 15000),
       seed=st.integers(0, 100))
@settings(max_examples=10, deadline=None)
def test_cc_uf_fuzz(scale, ne, seed):
    from lux_amd.cc_engine import CCUnionFindEngine
    full = DeviceCSC.rmat(scale, ne, seed=seed, sym=True)
    eng = CCUnionFindEngine(GraphPart(full, 1, 0))
    eng.run()
    got = eng.labels.cpu().numpy().view(np.uint32)
    want, _ = cpu_ref.cc(Graph.rmat(scale, ne, seed=seed, sym=True))
    np.testing.assert_array_equal(got, want)
