"""Reference converter CLI parity (tools/converter.cc surface:
`converter -nv N -ne M -input edges.txt -output g.lux`, README.md:56-75) —
CPU binary, runs in the no-GPU suite."""
import os
import subprocess

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _bin(name):
    path = os.path.join(ROOT, "bin", name)
    if not os.path.exists(path):
        import build
        build.build_apps()
    return path


def test_converter_roundtrip(tmp_path):
    from lux_amd.graph import Graph
    el = tmp_path / "e.txt"
    el.write_text("0 1\n1 2\n2 0\n2 1\n")
    out = str(tmp_path / "g.lux")
    r = subprocess.run([_bin("converter"), "-nv", "3", "-ne", "4", "-input",
                        str(el), "-output", out], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    g = Graph.load(out)
    assert (g.nv, g.ne) == (3, 4)
    # CSC by dst: dst0<-2, dst1<-{0,2}, dst2<-1
    np.testing.assert_array_equal(g.col_end, [1, 3, 4])
    assert sorted(g.src[1:3].tolist()) == [0, 2]
    assert g.src[0] == 2 and g.src[3] == 1


def test_rmat_gen_sym(tmp_path):
    """-sym emits both directions of ne/2 pairs, matching Graph.rmat
    sym=True (the components app's undirected input)."""
    import numpy as np

    from lux_amd.graph import Graph
    out = str(tmp_path / "s.lux")
    r = subprocess.run([_bin("rmat_gen"), "-kind", "rmat", "-scale", "9",
                        "-ne", "3000", "-sym", "-o", out],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    g = Graph.load(out)
    h = Graph.rmat(9, 3000, seed=1, sym=True)
    assert (g.nv, g.ne) == (h.nv, h.ne)
    np.testing.assert_array_equal(g.col_end, h.col_end)
    np.testing.assert_array_equal(np.sort(g.src), np.sort(h.src))
