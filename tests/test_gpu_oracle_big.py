"""Benchmark-scale correctness oracles, in-tree and repeatable
(the full-scale oracle runs existed only as BENCHLOG
lines). Env-gated: LUX_BIG_ORACLE=1 enables them (tens of GB of device
memory and a CPU-reference build per test — not for the default CI tier).

    gpurun -- 'LUX_BIG_ORACLE=1 python -m pytest tests/test_gpu_oracle_big.py -x -q'
"""
import os

import numpy as np
import pytest

pytestmark = [
    pytest.mark.gpu,
    pytest.mark.skipif(os.environ.get("LUX_BIG_ORACLE") != "1",
                       reason="env-gated (LUX_BIG_ORACLE=1): big-memory "
                              "benchmark-scale oracle runs"),
]


def test_sssp_rmat24_oracle_and_cpu_equivalence():
    """SSSP on RMAT-24 (16.8M V / 268M E): device check oracle reports 0
    violations AND the full label vector equals the C++ CPU reference."""
    from lux_amd.engine import DeviceCSC, GraphPart
    from lux_amd.graph import Graph
    from lux_amd.push_engine import PushEngine
    scale, ne, seed = 24, 1 << 28, 1
    eng = PushEngine(GraphPart(DeviceCSC.rmat(scale, ne, seed=seed), 1, 0),
                     PushEngine.MODE_MIN, source=0)
    iters = eng.run()
    assert iters > 0
    assert eng.check() == 0
    got = eng.labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed)
    from lux_amd import cpu_ref
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)


def test_cc_twitter_shaped_oracle():
    """Union-find CC on the Twitter-shaped symmetric synthetic
    (41.6M V / 200M E here — same generator family as BASELINE config 4):
    check oracle 0 violations; labels idempotent under the label-prop
    fixpoint property."""
    from lux_amd.cc_engine import CCUnionFindEngine
    from lux_amd.engine import DeviceCSC, GraphPart
    nv, ne, seed = 41652230, 200_000_000, 1
    full = DeviceCSC.rmat_folded(nv, ne, seed=seed, sym=True)
    eng = CCUnionFindEngine(GraphPart(full, 1, 0))
    eng.run()
    assert eng.check() == 0
