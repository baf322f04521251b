"""Native CLI app drivers (bin/*, single-GPU C++ runtime) smoke-tested on a
real GPU: the reference's one-binary-per-app surface (README.md:42-45 run
commands; ELAPSED TIME line pagerank.cc:118) served by the HIP runtime in
src/runtime/single_gpu.cpp."""
import os
import subprocess

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(ROOT, "bin")


def _run(args):
    r = subprocess.run(args, cwd=ROOT, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, f"{args}: {r.stdout}\n{r.stderr}"
    return r.stdout + r.stderr


def test_native_pagerank_synthetic():
    out = _run([f"{BIN}/pagerank", "-synthetic", "rmat:14:200000", "-ni",
                "5"])
    assert "ELAPSED TIME" in out


def test_native_sssp_check():
    out = _run([f"{BIN}/sssp", "-synthetic", "rmat:13:100000", "-start", "0",
                "-check"])
    assert "ELAPSED TIME" in out
    assert "PASS" in out


def test_native_components_check():
    out = _run([f"{BIN}/components", "-synthetic", "rmat:13:100000",
                "-check"])
    assert "ELAPSED TIME" in out
    assert "PASS" in out


def test_native_col_filter():
    out = _run([f"{BIN}/col_filter", "-synthetic",
                "bipartite:4000:500:100000", "-ni", "3", "-k", "32"])
    assert "ELAPSED TIME" in out


def test_native_lux_file_roundtrip(tmp_path):
    lux = str(tmp_path / "g.lux")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "13", "-ne",
          "100000", "-o", lux])
    out = _run([f"{BIN}/pagerank", "-file", lux, "-ni", "3", "-verbose"])
    assert "ELAPSED TIME" in out


def test_native_pagerank_dump_matches_cpu(tmp_path):
    """Cross-implementation parity: the native binary's -dump state equals
    the CPU reference PageRank on the same .lux file (the native PR path
    has no -check oracle; this is its numeric validation)."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "g.lux")
    out = str(tmp_path / "pr.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "13", "-ne",
          "120000", "-o", lux])
    _run([f"{BIN}/pagerank", "-file", lux, "-ni", "5", "-dump", out])
    got, it = ck.load_state(out)
    assert it == 5
    g = Graph.load(lux)
    want = cpu_ref.pagerank(g, 5)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-9)


def test_native_sssp_dump_matches_cpu(tmp_path):
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "g.lux")
    out = str(tmp_path / "l.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "12", "-ne",
          "80000", "-o", lux])
    _run([f"{BIN}/sssp", "-file", lux, "-start", "0", "-dump", out])
    got, _ = ck.load_state(out)
    g = Graph.load(lux)
    want, _ = cpu_ref.sssp(g, 0)
    np.testing.assert_array_equal(got, want)


def test_native_col_filter_dump_matches_cpu(tmp_path):
    """Native CF SGD numerics vs the CPU reference (this parity test is
    what exposed the missing new=old*(1-gamma*lambda) seed in the native
    runtime's sweep loop)."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "b.lux")
    out = str(tmp_path / "v.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "bipartite", "-users", "800", "-items",
          "120", "-ne", "20000", "-o", lux])
    _run([f"{BIN}/col_filter", "-file", lux, "-ni", "3", "-k", "20",
          "-dump", out])
    got, _ = ck.load_state(out)
    g = Graph.load(lux, want_weights=True)
    want = cpu_ref.cf(g, 20, 3)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-7)


def test_native_col_filter_als(tmp_path):
    """Native -als runs and reaches a lower loss than the same number of
    SGD sweeps."""
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "b.lux")
    o1, o2 = str(tmp_path / "sgd.luxs"), str(tmp_path / "als.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "bipartite", "-users", "2000", "-items",
          "300", "-ne", "60000", "-o", lux])
    _run([f"{BIN}/col_filter", "-file", lux, "-ni", "3", "-k", "32",
          "-dump", o1])
    # -users gives -als the bipartite boundary -> Gauss-Seidel alternation
    _run([f"{BIN}/col_filter", "-file", lux, "-ni", "3", "-k", "32", "-als",
          "-users", "2000", "-dump", o2])
    g = Graph.load(lux, want_weights=True)
    sgd, _ = ck.load_state(o1)
    als, _ = ck.load_state(o2)
    assert cpu_ref.cf_loss(g, 32, als) < cpu_ref.cf_loss(g, 32, sgd)


def test_native_components_uf_dump_matches_cpu(tmp_path):
    """Native union-find CC on an undirected graph: labels equal the CPU
    reference's converged max-label propagation."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "s.lux")
    out = str(tmp_path / "l.luxs")
    g = Graph.rmat(13, 60000, seed=3, sym=True)
    g.save(lux)
    o = _run([f"{BIN}/components", "-file", lux, "-dump", out, "-check"])
    assert "PASS" in o
    got, _ = ck.load_state(out)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(got, want)


def test_native_components_labelprop_dump(tmp_path):
    """-labelprop keeps the reference's directional propagation exactly
    (directed input)."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "d.lux")
    out = str(tmp_path / "l.luxs")
    g = Graph.rmat(12, 50000, seed=5)
    g.save(lux)
    _run([f"{BIN}/components", "-file", lux, "-labelprop", "-dump", out])
    got, _ = ck.load_state(out)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(got, want)


def test_native_pagerank_multi_rccl_world1(tmp_path):
    """The fork + RCCL native multi-GPU engine (run_pagerank_multi),
    exercised at world 1 on this box (LUX_NATIVE_MULTI=1): same numeric
    result as the CPU reference. The -ll:gpu N>1 path is this exact code
    with more children."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "g.lux")
    out = str(tmp_path / "pr.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "13", "-ne",
          "120000", "-o", lux])
    env = dict(os.environ, LUX_NATIVE_MULTI="1")
    r = subprocess.run([f"{BIN}/pagerank", "-file", lux, "-ni", "5",
                        "-dump", out], cwd=ROOT, capture_output=True,
                       text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "ELAPSED TIME" in r.stdout
    got, _ = ck.load_state(out)
    g = Graph.load(lux)
    want = cpu_ref.pagerank(g, 5)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-9)


def test_native_components_multi_rccl_world1(tmp_path):
    """Native multi-GPU union-find worker (components_multi_child) at
    world 1: same labelling as the CPU reference, and the star-exchange
    convergence loop terminates."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "s.lux")
    out = str(tmp_path / "l.luxs")
    g = Graph.rmat(13, 60000, seed=3, sym=True)
    g.save(lux)
    env = dict(os.environ, LUX_NATIVE_MULTI="1")
    r = subprocess.run([f"{BIN}/components", "-file", lux, "-dump", out,
                        "-check"], cwd=ROOT, capture_output=True, text=True,
                       timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PASS" in r.stdout
    got, _ = ck.load_state(out)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(got, want)


def test_native_col_filter_multi_rccl_world1(tmp_path):
    """Native multi-GPU CF worker (col_filter_multi_child) at world 1:
    SGD sweeps match the CPU reference; -als runs and beats SGD loss."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "b.lux")
    o1 = str(tmp_path / "v.luxs")
    o2 = str(tmp_path / "a.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "bipartite", "-users", "800",
          "-items", "120", "-ne", "20000", "-o", lux])
    env = dict(os.environ, LUX_NATIVE_MULTI="1")
    r = subprocess.run([f"{BIN}/col_filter", "-file", lux, "-ni", "3",
                        "-k", "20", "-dump", o1], cwd=ROOT,
                       capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    got, _ = ck.load_state(o1)
    g = Graph.load(lux, want_weights=True)
    want = cpu_ref.cf(g, 20, 3)
    np.testing.assert_allclose(got, want, rtol=2e-4, atol=1e-7)
    r = subprocess.run([f"{BIN}/col_filter", "-file", lux, "-ni", "3",
                        "-k", "32", "-als", "-users", "800", "-dump", o2],
                       cwd=ROOT,
                       capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    als, _ = ck.load_state(o2)
    sgd = cpu_ref.cf(g, 32, 3)
    assert cpu_ref.cf_loss(g, 32, als) < cpu_ref.cf_loss(g, 32, sgd)


def test_native_sssp_multi_rccl_world1(tmp_path):
    """Native multi-GPU push worker (push_multi_child) at world 1: exact
    SSSP labels + passing oracle through the meta-record exchange path."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "g.lux")
    out = str(tmp_path / "l.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "13", "-ne",
          "100000", "-o", lux])
    env = dict(os.environ, LUX_NATIVE_MULTI="1")
    r = subprocess.run([f"{BIN}/sssp", "-file", lux, "-start", "0",
                        "-check", "-dump", out], cwd=ROOT,
                       capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "PASS" in r.stdout
    got, _ = ck.load_state(out)
    g = Graph.load(lux)
    want, _ = cpu_ref.sssp(g, 0)
    np.testing.assert_array_equal(got, want)


def test_native_components_labelprop_multi_rccl_world1(tmp_path):
    """Native multi-GPU push worker in MODE_MAX (-labelprop) at world 1."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "d.lux")
    out = str(tmp_path / "l.luxs")
    g = Graph.rmat(12, 50000, seed=5)
    g.save(lux)
    env = dict(os.environ, LUX_NATIVE_MULTI="1")
    r = subprocess.run([f"{BIN}/components", "-file", lux, "-labelprop",
                        "-dump", out], cwd=ROOT, capture_output=True,
                       text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    got, _ = ck.load_state(out)
    want, _ = cpu_ref.cc(g)
    np.testing.assert_array_equal(got, want)


def test_native_blocked_pull_parity(tmp_path):
    """BlockedPull (the native src-blocked CSC) forced on at small scale
    (LUX_NATIVE_BLOCK_SHIFT=9): pagerank and sssp dumps must equal the
    CPU references — same numbers as the unblocked path."""
    import numpy as np
    from lux_amd import checkpoint as ck
    from lux_amd import cpu_ref
    from lux_amd.graph import Graph
    lux = str(tmp_path / "g.lux")
    o1 = str(tmp_path / "pr.luxs")
    o2 = str(tmp_path / "ss.luxs")
    _run([f"{BIN}/rmat_gen", "-kind", "rmat", "-scale", "13", "-ne",
          "120000", "-o", lux])
    env = dict(os.environ, LUX_NATIVE_BLOCK_SHIFT="9")
    for cmd, out in [([f"{BIN}/pagerank", "-file", lux, "-ni", "5",
                       "-dump", o1], o1),
                     ([f"{BIN}/sssp", "-file", lux, "-start", "0",
                       "-check", "-dump", o2], o2)]:
        r = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                           timeout=300, env=env)
        assert r.returncode == 0, r.stdout + r.stderr
    g = Graph.load(lux)
    got_pr, _ = ck.load_state(o1)
    np.testing.assert_allclose(got_pr, cpu_ref.pagerank(g, 5), rtol=2e-4,
                               atol=1e-9)
    got_ss, _ = ck.load_state(o2)
    want, _ = cpu_ref.sssp(g, 0)
    np.testing.assert_array_equal(got_ss, want)
