"""Union-find CC engine (src/gpu/cc_uf.hip) vs the CPU reference.

On SYMMETRIC graphs (connected components' standard input) the UF labelling
is IDENTICAL to converged max-label propagation (max vertex id per
component). On directed inputs label propagation computes directional
max-reachability instead — the components app keeps `-labelprop` for that
exact reference behavior, and these tests use sym=True graphs."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from lux_amd import _native_gpu as ng  # noqa: E402
from lux_amd import cpu_ref  # noqa: E402
from lux_amd.cc_engine import CCUnionFindEngine  # noqa: E402
from lux_amd.engine import DeviceCSC, GraphPart  # noqa: E402
from lux_amd.graph import Graph  # noqa: E402


def stream():
    return torch.cuda.current_stream().cuda_stream


@pytest.mark.parametrize("scale,ne,seed", [(12, 60000, 3), (14, 400000, 9),
                                           (10, 2000, 5)])
def test_cc_uf_matches_labelprop_reference(scale, ne, seed):
    full = DeviceCSC.rmat(scale, ne, seed=seed, sym=True)
    eng = CCUnionFindEngine(GraphPart(full, 1, 0))
    eng.run()
    got = eng.labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed, sym=True)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(got, want)
    assert eng.check() == 0


def test_cc_uf_multipart_star_exchange():
    """2 partitions with the engine's star-forest exchange emulated in one
    process: each rank unions only its own edges, then peers' label vectors
    are unioned as stars until stable — must match the whole-graph result."""
    scale, ne, seed = 13, 150000, 21
    nv = 1 << scale
    full = DeviceCSC.rmat(scale, ne, seed=seed, sym=True)
    pa = GraphPart(full, 2, 0, keep_full=True)
    pb = GraphPart(full, 2, 1)
    ea, eb = CCUnionFindEngine(pa), CCUnionFindEngine(pb)
    ea.run()
    eb.run()
    s = stream()
    for _ in range(6):
        la = ea.labels_t.clone()
        lb = eb.labels_t.clone()
        ng.uf_union_star(s, nv, lb, ea.parent)
        ng.uf_union_star(s, nv, la, eb.parent)
        ng.uf_flatten(s, nv, ea.parent, ea.labels_t)
        ng.uf_flatten(s, nv, eb.parent, eb.labels_t)
        if bool((ea.labels_t == la).all()) and bool(
                (eb.labels_t == lb).all()):
            break
    g = Graph.rmat(scale, ne, seed=seed, sym=True)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(ea.labels.cpu().numpy().view(np.uint32),
                                  want)
    np.testing.assert_array_equal(eb.labels.cpu().numpy().view(np.uint32),
                                  want)
