"""ctypes bindings for the native lux libraries.

Two libraries, both built in-tree by build.py:
  liblux_cpu.so — always available (g++); .lux IO, RMAT, partitioner, CPU
                  reference engines.
  liblux_gpu.so — gfx950 HIP kernels (hipcc); loads on any box with the ROCm
                  runtime installed, callable only where a GPU exists.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))

u32p = ctypes.POINTER(ctypes.c_uint32)
u64p = ctypes.POINTER(ctypes.c_uint64)
i32p = ctypes.POINTER(ctypes.c_int32)
f32p = ctypes.POINTER(ctypes.c_float)


def _load(name):
    path = os.path.join(_DIR, name)
    if not os.path.exists(path):
        return None
    return ctypes.CDLL(path)


def _build_and_load_cpu():
    lib = _load("liblux_cpu.so")
    if lib is None:
        import build  # noqa: repo-root build.py (repo root on sys.path)
        build.build_cpu()
        lib = _load("liblux_cpu.so")
    return lib


def ptr(arr, ctype):
    """numpy array -> ctypes pointer (no copy); None passes NULL."""
    if arr is None:
        return None
    assert arr.flags["C_CONTIGUOUS"], "array must be C-contiguous"
    return arr.ctypes.data_as(ctype)


try:
    cpu = _build_and_load_cpu()
except Exception:  # pragma: no cover - build envs without a compiler
    cpu = None

gpu = _load("liblux_gpu.so")

if cpu is not None:
    cpu.lux_cf_loss.restype = ctypes.c_double
    cpu.lux_sssp_check.restype = ctypes.c_uint64
    cpu.lux_cc_check.restype = ctypes.c_uint64
    cpu.lux_sssp_iter_part.restype = ctypes.c_uint32
    cpu.lux_cc_iter_part.restype = ctypes.c_uint32
    cpu.lux_sssp_cpu.restype = ctypes.c_int
    cpu.lux_cc_cpu.restype = ctypes.c_int


# ---------------- CPU wrappers ----------------

def rmat_edges(seed, scale, ne):
    src = np.empty(ne, np.uint32)
    dst = np.empty(ne, np.uint32)
    cpu.lux_rmat_edges(ctypes.c_uint64(seed), ctypes.c_int(scale),
                       ctypes.c_uint64(ne), ptr(src, u32p), ptr(dst, u32p))
    return src, dst


def rmat_edges_folded(seed, scale, nv, ne):
    src = np.empty(ne, np.uint32)
    dst = np.empty(ne, np.uint32)
    cpu.lux_rmat_edges_folded(ctypes.c_uint64(seed), ctypes.c_int(scale),
                              ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                              ptr(src, u32p), ptr(dst, u32p))
    return src, dst


def bipartite_edges(seed, n_users, n_items, ne):
    src = np.empty(ne, np.uint32)
    dst = np.empty(ne, np.uint32)
    w = np.empty(ne, np.int32)
    cpu.lux_bipartite_edges(ctypes.c_uint64(seed), ctypes.c_uint32(n_users),
                            ctypes.c_uint32(n_items), ctypes.c_uint64(ne),
                            ptr(src, u32p), ptr(dst, u32p), ptr(w, i32p))
    return src, dst, w


def edges_to_csc(nv, src, dst, w=None):
    ne = len(src)
    col_end = np.empty(nv, np.uint64)
    out_src = np.empty(ne, np.uint32)
    out_w = np.empty(ne, np.int32) if w is not None else None
    cpu.lux_edges_to_csc(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                         ptr(src, u32p), ptr(dst, u32p), ptr(w, i32p),
                         ptr(col_end, u64p), ptr(out_src, u32p),
                         ptr(out_w, i32p))
    return col_end, out_src, out_w


def io_write(path, nv, ne, col_end, src, weight=None):
    r = cpu.lux_io_write(path.encode(), ctypes.c_uint32(nv),
                         ctypes.c_uint64(ne), ptr(col_end, u64p),
                         ptr(src, u32p), ptr(weight, i32p))
    if r != 0:
        raise IOError(f"lux_write failed: {path}")


def io_read_header(path):
    nv = ctypes.c_uint32()
    ne = ctypes.c_uint64()
    w = ctypes.c_int()
    r = cpu.lux_io_read_header(path.encode(), ctypes.byref(nv),
                               ctypes.byref(ne), ctypes.byref(w))
    if r != 0:
        raise IOError(f"lux_read_header failed: {path}")
    return nv.value, ne.value, bool(w.value)


def io_read(path, want_weights=False):
    nv, ne, weighted = io_read_header(path)
    col_end = np.empty(nv, np.uint64)
    src = np.empty(ne, np.uint32)
    weight = None
    if want_weights:
        if not weighted:
            raise IOError(f"{path} has no weights")
        weight = np.empty(ne, np.int32)
    r = cpu.lux_io_read(path.encode(), ptr(col_end, u64p), ptr(src, u32p),
                        ptr(weight, i32p))
    if r != 0:
        raise IOError(f"lux_read failed: {path}")
    return nv, ne, col_end, src, weight


def io_read_col_end(path, nv):
    col_end = np.empty(nv, np.uint64)
    r = cpu.lux_io_read_col_end(path.encode(), ptr(col_end, u64p))
    if r != 0:
        raise IOError(f"lux_read_col_end failed: {path}")
    return col_end


def io_read_slice(path, row_left, row_right, ep, want_weights=False):
    """Read ONLY [row_left, row_right]'s col_end + edge slice (ep edges,
    known from a prior col_end read). Reference parity: the per-partition
    fseeko load (core/push_model.inl:100-119)."""
    nrows = row_right - row_left + 1
    col_end = np.empty(nrows, np.uint64)
    src = np.empty(max(ep, 1), np.uint32)
    weight = np.empty(max(ep, 1), np.int32) if want_weights else None
    cpu.lux_io_read_slice.restype = ctypes.c_int64
    r = cpu.lux_io_read_slice(path.encode(), ctypes.c_uint32(row_left),
                              ctypes.c_uint32(row_right),
                              ptr(col_end, u64p), ptr(src, u32p),
                              ptr(weight, i32p))
    if r < 0:
        raise IOError(f"lux_read_slice failed: {path}")
    if r != ep:
        raise IOError(f"lux_read_slice: {path}: got {r} edges, want {ep}")
    return col_end, src[:ep], (weight[:ep] if weight is not None else None)


def partition(nv, ne, col_end, nparts):
    rl = np.empty(nparts, np.uint32)
    rr = np.empty(nparts, np.uint32)
    cl = np.empty(nparts, np.uint64)
    cr = np.empty(nparts, np.uint64)
    cpu.lux_partition(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                      ptr(col_end, u64p), ctypes.c_int(nparts),
                      ptr(rl, u32p), ptr(rr, u32p), ptr(cl, u64p),
                      ptr(cr, u64p))
    return rl, rr, cl, cr


def out_degrees(nv, src):
    deg = np.empty(nv, np.uint32)
    cpu.lux_out_degrees(ctypes.c_uint32(nv), ctypes.c_uint64(len(src)),
                        ptr(src, u32p), ptr(deg, u32p))
    return deg


def pagerank_init(nv, deg):
    pr = np.empty(nv, np.float32)
    cpu.lux_pagerank_init(ctypes.c_uint32(nv), ptr(deg, u32p), ptr(pr, f32p))
    return pr


def pagerank_iter_part(nv, row_left, row_right, col_left, col_end_slice,
                       src_slice, deg, old_pr, new_pr_part):
    cpu.lux_pagerank_iter_part(
        ctypes.c_uint32(nv), ctypes.c_uint32(row_left),
        ctypes.c_uint32(row_right), ctypes.c_uint64(col_left),
        ptr(col_end_slice, u64p), ptr(src_slice, u32p), ptr(deg, u32p),
        ptr(old_pr, f32p), ptr(new_pr_part, f32p))


def pagerank_cpu(nv, ne, col_end, src, iters):
    pr = np.empty(nv, np.float32)
    cpu.lux_pagerank_cpu(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                         ptr(col_end, u64p), ptr(src, u32p),
                         ctypes.c_int(iters), ptr(pr, f32p))
    return pr


def sssp_iter_part(row_left, row_right, col_left, col_end_slice, src_slice,
                   old_label, new_label_part):
    return cpu.lux_sssp_iter_part(
        ctypes.c_uint32(row_left), ctypes.c_uint32(row_right),
        ctypes.c_uint64(col_left), ptr(col_end_slice, u64p),
        ptr(src_slice, u32p), ptr(old_label, u32p), ptr(new_label_part, u32p))


def cc_iter_part(row_left, row_right, col_left, col_end_slice, src_slice,
                 old_label, new_label_part):
    return cpu.lux_cc_iter_part(
        ctypes.c_uint32(row_left), ctypes.c_uint32(row_right),
        ctypes.c_uint64(col_left), ptr(col_end_slice, u64p),
        ptr(src_slice, u32p), ptr(old_label, u32p), ptr(new_label_part, u32p))


def sssp_cpu(nv, ne, col_end, src, source):
    label = np.empty(nv, np.uint32)
    iters = cpu.lux_sssp_cpu(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                             ptr(col_end, u64p), ptr(src, u32p),
                             ctypes.c_uint32(source), ptr(label, u32p))
    return label, iters


def cc_cpu(nv, ne, col_end, src):
    label = np.empty(nv, np.uint32)
    iters = cpu.lux_cc_cpu(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                           ptr(col_end, u64p), ptr(src, u32p),
                           ptr(label, u32p))
    return label, iters


def sssp_check(nv, ne, col_end, src, label):
    return cpu.lux_sssp_check(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                              ptr(col_end, u64p), ptr(src, u32p),
                              ptr(label, u32p))


def cc_check(nv, ne, col_end, src, label):
    return cpu.lux_cc_check(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                            ptr(col_end, u64p), ptr(src, u32p),
                            ptr(label, u32p))


def cf_init(nv, K):
    vec = np.empty((nv, K), np.float32)
    cpu.lux_cf_init(ctypes.c_uint32(nv), ctypes.c_int(K), ptr(vec, f32p))
    return vec


def cf_iter_part(row_left, row_right, col_left, col_end_slice, src_slice,
                 w_slice, K, old_vec, new_vec_part):
    cpu.lux_cf_iter_part(
        ctypes.c_uint32(row_left), ctypes.c_uint32(row_right),
        ctypes.c_uint64(col_left), ptr(col_end_slice, u64p),
        ptr(src_slice, u32p), ptr(w_slice, i32p), ctypes.c_int(K),
        ptr(old_vec, f32p), ptr(new_vec_part, f32p))


def cf_cpu(nv, ne, col_end, src, w, K, iters):
    vec = np.empty((nv, K), np.float32)
    cpu.lux_cf_cpu(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                   ptr(col_end, u64p), ptr(src, u32p), ptr(w, i32p),
                   ctypes.c_int(K), ctypes.c_int(iters), ptr(vec, f32p))
    return vec


def cf_loss(nv, ne, col_end, src, w, K, vec):
    return cpu.lux_cf_loss(ctypes.c_uint32(nv), ctypes.c_uint64(ne),
                           ptr(col_end, u64p), ptr(src, u32p), ptr(w, i32p),
                           ctypes.c_int(K), ptr(vec, f32p))
