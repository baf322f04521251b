"""Frontier format constants: python (lux_amd/types.py) vs the C++ header
(src/include/lux/types.h) vs the reference's literal values
(core/graph.h:100-106, core/push_model.inl:391-412)."""
import os
import re

from lux_amd import types as t

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _header():
    return open(os.path.join(ROOT, "src", "include", "lux", "types.h")).read()


def test_constants_match_header():
    h = _header()
    assert hex(t.DENSE_BITMAP) in h.replace("u;", ";")
    assert hex(t.SPARSE_QUEUE) in h.replace("u;", ";")
    assert f"SPARSE_THRESHOLD = {t.SPARSE_THRESHOLD}" in h
    assert f"SLIDING_WINDOW = {t.SLIDING_WINDOW}" in h
    assert "0xFFFFFFFF" in h  # INF_LABEL
    assert "0.15f" in h and "0.001f" in h and "0.00000035f" in h


def test_reference_literals():
    # byte-compatibility with the reference header tags (core/graph.h:103-104)
    assert t.DENSE_BITMAP == 0x1234567
    assert t.SPARSE_QUEUE == 0x7654321
    # sizing rule (core/push_model.inl:391-412): (verts/16+100)*4 + 8 hdr
    assert t.frontier_bytes(1600) == (1600 // 16 + 100) * 4 + 8
    assert t.frontier_capacity(1600) == 1600 // 16 + 100
