"""Shared constants, mirroring src/include/lux/types.h (kept in sync by
tests/test_frontier.py)."""

DENSE_BITMAP = 0x1234567
SPARSE_QUEUE = 0x7654321
SPARSE_THRESHOLD = 16
SLIDING_WINDOW = 4
INF_LABEL = 0xFFFFFFFF
PR_ALPHA = 0.15
CF_LAMBDA = 0.001
CF_GAMMA = 0.00000035


def frontier_bytes(part_verts: int) -> int:
    """Frontier segment byte size for a partition (reference rule,
    core/push_model.inl:391-412)."""
    return (part_verts // SPARSE_THRESHOLD + 100) * 4 + 8


def frontier_capacity(part_verts: int) -> int:
    """Max sparse-queue entries in a segment (sssp_gpu.cu:410)."""
    return part_verts // SPARSE_THRESHOLD + 100
