"""Python app drivers (lux_amd.apps.*) exercised end-to-end via their
main() on small synthetic graphs — CLI parsing, memory estimate, engine
run, ELAPSED TIME line, -check oracles, -verbose, -als, -labelprop."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_pagerank_app_main(capsys):
    from lux_amd.apps import pagerank
    eng = pagerank.main(["-synthetic", "rmat:12:100000", "-ni", "3",
                         "-verbose"])
    out = capsys.readouterr().out
    assert "ELAPSED TIME" in out
    assert np.isfinite(eng.ranks().cpu().numpy()).all()


def test_sssp_app_main_check(capsys):
    from lux_amd.apps import sssp
    sssp.main(["-synthetic", "rmat:12:80000", "-start", "0", "-check"])
    out = capsys.readouterr().out
    assert "ELAPSED TIME" in out
    assert "PASS" in out


def test_cc_app_main_check(capsys):
    from lux_amd.apps import cc
    cc.main(["-synthetic", "rmat:12:80000", "-check"])
    out = capsys.readouterr().out
    assert "ELAPSED TIME" in out
    assert "PASS" in out


def test_cc_app_labelprop(capsys):
    from lux_amd.apps import cc
    cc.main(["-synthetic", "rmat:11:40000", "-labelprop", "-check"])
    out = capsys.readouterr().out
    assert "PASS" in out


@pytest.mark.parametrize("extra", [[], ["-als"]])
def test_cf_app_main(capsys, extra):
    from lux_amd.apps import cf
    eng = cf.main(["-synthetic", "bipartite:2000:300:50000", "-ni", "2",
                   "-k", "32"] + extra)
    out = capsys.readouterr().out
    assert "ELAPSED TIME" in out
    assert np.isfinite(eng.vectors().cpu().numpy()).all()
