"""ctypes bindings for liblux_gpu.so (gfx950 HIP kernels).

All device pointers are passed as raw integers (torch `Tensor.data_ptr()`),
streams as `torch.cuda.current_stream().cuda_stream`. Dtype mapping:
u32 buffers <-> torch.int32, u64/E_ID <-> torch.int64, f32 <-> torch.float32
(bit-compatible; the native side interprets unsigned).

Every entry point raises if the library is missing — GPU paths must fail
loudly rather than fall back to eager/CPU silently.
"""
import ctypes
import os

_DIR = os.path.dirname(os.path.abspath(__file__))
_PATH = os.path.join(_DIR, "liblux_gpu.so")

_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_PATH):
            raise RuntimeError(
                "liblux_gpu.so not built — run `python build.py --gpu` "
                "(the HIP extension is mandatory on GPU boxes; no fallback)")
        _lib = ctypes.CDLL(_PATH)
        _lib.lux_gpu_scan_partials_size.restype = ctypes.c_uint32
    return _lib


def dp(t):
    """torch tensor -> device pointer (as c_void_p)."""
    if t is None:
        return ctypes.c_void_p(0)
    return ctypes.c_void_p(t.data_ptr())


def _u64(x):
    return ctypes.c_uint64(int(x))


def _u32(x):
    return ctypes.c_uint32(int(x))


def rmat_edges(stream, seed, scale, ne, src, dst):
    lib().lux_gpu_rmat_edges(_u64(stream), _u64(seed), ctypes.c_int(scale),
                             _u64(ne), dp(src), dp(dst))


def bipartite_edges(stream, seed, n_users, n_items, ne, src, dst, w):
    lib().lux_gpu_bipartite_edges(_u64(stream), _u64(seed), _u32(n_users),
                                  _u32(n_items), _u64(ne), dp(src), dp(dst),
                                  dp(w))


def hist_u32(stream, n, ids, hist):
    lib().lux_gpu_hist_u32(_u64(stream), _u64(n), dp(ids), dp(hist))


def scan_partials_size(n):
    return int(lib().lux_gpu_scan_partials_size(_u32(n)))


def scan_end_offsets(stream, n, inp, out_end, partials):
    lib().lux_gpu_scan_end_offsets(_u64(stream), _u32(n), dp(inp),
                                   dp(out_end), dp(partials))


def edges_to_csc(stream, nv, ne, src, dst, w, col_end, out_src, out_w, hist,
                 cursor, partials):
    lib().lux_gpu_edges_to_csc(_u64(stream), _u32(nv), _u64(ne), dp(src),
                               dp(dst), dp(w), dp(col_end), dp(out_src),
                               dp(out_w), dp(hist), dp(cursor), dp(partials))


def local_row_ptr(stream, vp, col_left, col_end_slice, row_ptr_loc):
    lib().lux_gpu_local_row_ptr(_u64(stream), _u32(vp), _u64(col_left),
                                dp(col_end_slice), dp(row_ptr_loc))


def build_bins(stream, vp, row_ptr, bin0, bin1, bin2, bin2v, counters):
    lib().lux_gpu_build_bins(_u64(stream), _u32(vp), dp(row_ptr), dp(bin0),
                             dp(bin1), dp(bin2), dp(bin2v), dp(counters))


PULL_PR = 0
PULL_MIN = 1
PULL_MAX = 2


def pull_iter(stream, mode, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v,
              row_ptr, col, oldv, newv, deg, row_left, init_rank):
    lib().lux_gpu_pull_iter(_u64(stream), ctypes.c_int(mode), _u32(n0),
                            dp(bin0), _u32(n1), dp(bin1), _u32(n2), dp(bin2),
                            _u32(nbig), dp(bin2v), dp(row_ptr), dp(col),
                            dp(oldv), dp(newv), dp(deg), _u32(row_left),
                            ctypes.c_float(init_rank))
