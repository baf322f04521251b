"""GPU push engine tests: SSSP/CC vs CPU golden, adaptive frontier paths,
check oracles, multi-partition composition in one process."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from lux_amd import cpu_ref  # noqa: E402
from lux_amd.engine import DeviceCSC, GraphPart  # noqa: E402
from lux_amd.graph import Graph  # noqa: E402
from lux_amd.push_engine import PushEngine  # noqa: E402
from lux_amd.types import DENSE_BITMAP, SPARSE_QUEUE  # noqa: E402

U32 = torch.int32


def run_single(scale, ne, seed, mode, source=0):
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    part = GraphPart(full, 1, 0)
    eng = PushEngine(part, mode, source=source)
    iters = eng.run(max_iters=1 << scale)
    return eng, iters


@pytest.mark.parametrize("scale,ne,seed", [(10, 8000, 3), (12, 300000, 5)])
def test_sssp_vs_cpu(scale, ne, seed):
    eng, _ = run_single(scale, ne, seed, PushEngine.MODE_MIN)
    got = eng.labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)
    assert eng.check() == 0


@pytest.mark.parametrize("scale,ne,seed", [(10, 8000, 7), (12, 300000, 9)])
def test_cc_vs_cpu(scale, ne, seed):
    eng, _ = run_single(scale, ne, seed, PushEngine.MODE_MAX)
    got = eng.labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed)
    want, _ = cpu_ref.cc(g)
    assert np.array_equal(got, want)
    assert eng.check() == 0


def test_sssp_sparse_path_used():
    """From a single source the early frontier must stay sparse (the
    reference's SPARSE_QUEUE path), only possibly densifying later."""
    full = DeviceCSC.rmat(12, 100000, seed=11)
    part = GraphPart(full, 1, 0)
    eng = PushEngine(part, PushEngine.MODE_MIN, source=0)
    assert eng.headers[0][0] == SPARSE_QUEUE and eng.headers[0][1] == 1
    eng.step()
    assert eng.headers[0][0] in (SPARSE_QUEUE, DENSE_BITMAP)


def test_cc_starts_dense_pull_fallback():
    """CC seeds an all-ones dense frontier (components_gpu.cu:733-740), so
    iteration 1 must take the pull fallback (nv > nv/16)."""
    full = DeviceCSC.rmat(10, 20000, seed=13)
    part = GraphPart(full, 1, 0)
    eng = PushEngine(part, PushEngine.MODE_MAX)
    assert eng.headers[0] == (DENSE_BITMAP, 1 << 10)
    eng.run(max_iters=2000)
    got = eng.labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(10, 20000, seed=13)
    want, _ = cpu_ref.cc(g)
    assert np.array_equal(got, want)


def test_sssp_multipart_single_process():
    """4 partitions in one process with manual segment/label cross-copy
    (the N-GPU equivalence shape of SURVEY.md §4(d) on one device)."""
    scale, ne, seed, P = 11, 60000, 17, 4
    nv = 1 << scale
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    parts = [GraphPart(full, P, p, keep_full=True) for p in range(P)]
    engines = [PushEngine(parts[p], PushEngine.MODE_MIN, source=0)
               for p in range(P)]
    for it in range(4 * nv):
        for e in engines:
            e.step()
        # manual exchange across the P single-rank engines: segments,
        # label annexes, label slices and the meta records
        mh = np.stack([e.meta_mine.cpu().numpy().view(np.uint32)
                       for e in engines])
        for ei in engines:
            for j, ej in enumerate(engines):
                if parts[j].vp == 0:
                    continue
                ei.labels.narrow(0, parts[j].row_left,
                                 parts[j].vp).copy_(ej.labels_part)
                ei.fq_all.narrow(0, int(ei.seg_off[j]),
                                 ei.seg_bytes[j]).copy_(ej.new_seg)
                cap_j = int(ei.annex_off[j + 1] - ei.annex_off[j])
                ei.fq_annex_all.narrow(0, int(ei.annex_off[j]),
                                       cap_j).copy_(ej.new_annex[:cap_j])
            ei.meta_host = mh.copy()
            ei.labels_current = True
            ei.headers = [(int(mh[q, 0]), int(mh[q, 1])) for q in range(P)]
        if int(mh[:, 1].sum()) == 0 and not mh[:, 4].any():
            break
    got = engines[0].labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)


def test_sssp_unreachable_stays_inf():
    # two disconnected cliques; source in first
    src = [0, 1, 2, 4, 5, 6]
    dst = [1, 2, 0, 5, 6, 4]
    g = Graph.from_edges(8, src, dst)
    gg = DeviceCSC.from_host(g)
    part = GraphPart(gg, 1, 0)
    eng = PushEngine(part, PushEngine.MODE_MIN, source=0)
    eng.run(max_iters=64)
    got = eng.labels.cpu().numpy().view(np.uint32)
    want, _ = cpu_ref.sssp(g, 0)
    assert np.array_equal(got, want)
    assert got[4] == 0xFFFFFFFF


def test_cc_labelprop_multipart_single_process():
    """MODE_MAX (CC label propagation) across 3 partitions with the manual
    meta/annex exchange: unlike BFS, queued labels here are arbitrary
    values, so this exercises the annex-fill-after-fixup path (labels
    must be iteration-final, not enqueue-time)."""
    scale, ne, seed, P = 10, 30000, 29, 3
    nv = 1 << scale
    full = DeviceCSC.rmat(scale, ne, seed=seed)
    parts = [GraphPart(full, P, p, keep_full=True) for p in range(P)]
    engines = [PushEngine(parts[p], PushEngine.MODE_MAX) for p in range(P)]
    for it in range(4 * nv):
        for e in engines:
            e.step()
        mh = np.stack([e.meta_mine.cpu().numpy().view(np.uint32)
                       for e in engines])
        for ei in engines:
            for j, ej in enumerate(engines):
                if parts[j].vp == 0:
                    continue
                ei.labels.narrow(0, parts[j].row_left,
                                 parts[j].vp).copy_(ej.labels_part)
                ei.fq_all.narrow(0, int(ei.seg_off[j]),
                                 ei.seg_bytes[j]).copy_(ej.new_seg)
                cap_j = int(ei.annex_off[j + 1] - ei.annex_off[j])
                ei.fq_annex_all.narrow(0, int(ei.annex_off[j]),
                                       cap_j).copy_(ej.new_annex[:cap_j])
            ei.meta_host = mh.copy()
            ei.labels_current = True
            ei.headers = [(int(mh[q, 0]), int(mh[q, 1])) for q in range(P)]
        if int(mh[:, 1].sum()) == 0 and not mh[:, 4].any():
            break
    got = engines[0].labels.cpu().numpy().view(np.uint32)
    g = Graph.rmat(scale, ne, seed=seed)
    want, _ = cpu_ref.cc(g)
    assert np.array_equal(got, want)
