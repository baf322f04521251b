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
    """torch tensor (or raw int device address) -> c_void_p."""
    if t is None:
        return ctypes.c_void_p(0)
    if isinstance(t, int):
        return ctypes.c_void_p(t)
    return ctypes.c_void_p(t.data_ptr())


def _u64(x):
    return ctypes.c_uint64(int(x))


def _u32(x):
    return ctypes.c_uint32(int(x))


def rmat_edges(stream, seed, scale, ne, src, dst):
    lib().lux_gpu_rmat_edges(_u64(stream), _u64(seed), ctypes.c_int(scale),
                             _u64(ne), dp(src), dp(dst))


def rmat_edges_chunk(stream, seed, scale, e0, ne, src, dst):
    lib().lux_gpu_rmat_edges_chunk(_u64(stream), _u64(seed),
                                   ctypes.c_int(scale), _u64(e0), _u64(ne),
                                   dp(src), dp(dst))


def rmat_edges_folded(stream, seed, scale, nv, ne, src, dst):
    lib().lux_gpu_rmat_edges_folded(_u64(stream), _u64(seed),
                                    ctypes.c_int(scale), _u32(nv), _u64(ne),
                                    dp(src), dp(dst))


def rmat_edges_folded_chunk(stream, seed, scale, nv, e0, ne, src, dst):
    lib().lux_gpu_rmat_edges_folded_chunk(
        _u64(stream), _u64(seed), ctypes.c_int(scale), _u32(nv), _u64(e0),
        _u64(ne), dp(src), dp(dst))


def bipartite_edges_chunk(stream, seed, n_users, n_items, e0, ne, src, dst,
                          w):
    lib().lux_gpu_bipartite_edges_chunk(
        _u64(stream), _u64(seed), _u32(n_users), _u32(n_items), _u64(e0),
        _u64(ne), dp(src), dp(dst), dp(w))


def slice_scatter(stream, n, src, dst, w, rl, rr, cursor, out_col, out_w):
    lib().lux_gpu_slice_scatter(_u64(stream), _u64(n), dp(src), dp(dst),
                                dp(w), _u32(rl), _u32(rr), dp(cursor),
                                dp(out_col), dp(out_w))


def cf_iter(stream, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v, row_ptr, col,
            w, oldv, newv, row_left, K):
    lib().lux_gpu_cf_iter(_u64(stream), _u32(n0), dp(bin0), _u32(n1),
                          dp(bin1), _u32(n2), dp(bin2), _u32(nbig), dp(bin2v),
                          dp(row_ptr), dp(col), dp(w), dp(oldv), dp(newv),
                          _u32(row_left), ctypes.c_int(K))


def bipartite_edges(stream, seed, n_users, n_items, ne, src, dst, w):
    lib().lux_gpu_bipartite_edges(_u64(stream), _u64(seed), _u32(n_users),
                                  _u32(n_items), _u64(ne), dp(src), dp(dst),
                                  dp(w))


def hist_u32(stream, n, ids, hist):
    lib().lux_gpu_hist_u32(_u64(stream), _u64(n), dp(ids), dp(hist))


def scan_partials_size(n):
    return int(lib().lux_gpu_scan_partials_size(_u64(n)))


def scan_end_offsets(stream, n, inp, out_end, partials):
    lib().lux_gpu_scan_end_offsets(_u64(stream), _u64(n), dp(inp),
                                   dp(out_end), dp(partials))


def u64_to_u32(stream, n, inp, out):
    lib().lux_gpu_u64_to_u32(_u64(stream), _u64(n), dp(inp), dp(out))


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


def csr_scatter(stream, ep, col, row_ptr_loc, vp, row_left, cursor,
                push_col):
    lib().lux_gpu_csr_scatter(_u64(stream), _u64(ep), dp(col),
                              dp(row_ptr_loc), _u32(vp), _u32(row_left),
                              dp(cursor), dp(push_col))


def build_bitmap(stream, vp, snapshot, new_labels, seg):
    lib().lux_gpu_build_bitmap(_u64(stream), _u32(vp), dp(snapshot),
                               dp(new_labels), dp(seg))


def d2s(stream, vp, row_left, dense_seg, sparse_seg):
    lib().lux_gpu_d2s(_u64(stream), _u32(vp), _u32(row_left), dp(dense_seg),
                      dp(sparse_seg))


def check(stream, is_min, vp, row_left, row_ptr_loc, col, labels, mistakes):
    lib().lux_gpu_check(_u64(stream), ctypes.c_int(is_min), _u32(vp),
                        _u32(row_left), dp(row_ptr_loc), dp(col), dp(labels),
                        dp(mistakes))


PULL_PR = 0
PULL_MIN = 1
PULL_MAX = 2


def pull_iter(stream, mode, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v,
              row_ptr, col, oldv, newv, deg, row_left, init_rank,
              row_u32=0):
    lib().lux_gpu_pull_iter(_u64(stream), ctypes.c_int(mode), _u32(n0),
                            dp(bin0), _u32(n1), dp(bin1), _u32(n2), dp(bin2),
                            _u32(nbig), dp(bin2v), dp(row_ptr),
                            ctypes.c_int(row_u32), dp(col),
                            dp(oldv), dp(newv), dp(deg), _u32(row_left),
                            ctypes.c_float(init_rank))


def blocked_count(stream, ep, col, row_ptr_loc, vp, bounds, nb, counts,
                  lo=0, hi=0xFFFFFFFF):
    lib().lux_gpu_blocked_count(_u64(stream), _u64(ep), dp(col),
                                dp(row_ptr_loc), _u32(vp), dp(bounds),
                                ctypes.c_int(nb), _u32(lo), _u32(hi),
                                dp(counts))


def blocked_scatter(stream, ep, col, row_ptr_loc, vp, bounds, nb, cursor,
                    out_col, lo=0, hi=0xFFFFFFFF):
    lib().lux_gpu_blocked_scatter(_u64(stream), _u64(ep), dp(col),
                                  dp(row_ptr_loc), _u32(vp), dp(bounds),
                                  ctypes.c_int(nb), _u32(lo), _u32(hi),
                                  dp(cursor), dp(out_col))


def pull_finish_pr(stream, vp, newv, deg, row_left, init_rank):
    lib().lux_gpu_pull_finish_pr(_u64(stream), _u32(vp), dp(newv), dp(deg),
                                 _u32(row_left), ctypes.c_float(init_rank))


def cf_als_iter(stream, n0, bin0, n1, bin1, n2, bin2, nbig, bin2v, hubidx,
                gram_scratch, rhs_scratch, row_ptr, col, w, oldv, newv,
                row_left, K, oldv_bf=None):
    lib().lux_gpu_cf_als_iter(
        _u64(stream), _u32(n0), dp(bin0), _u32(n1), dp(bin1), _u32(n2),
        dp(bin2), _u32(nbig), dp(bin2v), dp(hubidx), dp(gram_scratch),
        dp(rhs_scratch), dp(row_ptr), dp(col), dp(w), dp(oldv), dp(oldv_bf),
        dp(newv), _u32(row_left), ctypes.c_int(K))


def frontier_expand(stream, old_dense, in_row_left, in_count, old_seg,
                    qlabels, labels_repair, push_row_ptr, items, counter,
                    max_items):
    lib().lux_gpu_frontier_expand(
        _u64(stream), ctypes.c_int(old_dense), _u32(in_row_left),
        _u32(in_count), dp(old_seg), dp(qlabels), dp(labels_repair),
        dp(push_row_ptr), dp(items), dp(counter), _u32(max_items))


def frontier_expand_auto(stream, verts, in_row_left, seg, qlabels,
                         labels_repair, push_row_ptr, items, counter,
                         max_items):
    lib().lux_gpu_frontier_expand_auto(
        _u64(stream), _u32(verts), _u32(in_row_left), dp(seg), dp(qlabels),
        dp(labels_repair), dp(push_row_ptr), dp(items), dp(counter),
        _u32(max_items))


def frontier_fixup(stream, vp, row_left, capacity, built_dense, snapshot,
                   labels_part, deg_part, new_seg, annex, tmp_seg, meta,
                   item_counter, max_items):
    """Device-predicated conversion chain + meta record (push.hip
    lux_gpu_frontier_fixup): zero host syncs."""
    lib().lux_gpu_frontier_fixup(
        _u64(stream), _u32(vp), _u32(row_left), _u32(capacity),
        ctypes.c_int(built_dense), dp(snapshot), dp(labels_part),
        dp(deg_part), dp(new_seg), dp(annex), dp(tmp_seg), dp(meta),
        dp(item_counter), _u32(max_items))


def push_chunk_scatter(stream, is_min, new_dense, items, counter, max_items,
                       push_row_ptr, push_col, old_labels, snapshot,
                       new_labels, my_row_left, new_seg, capacity,
                       visited_bits=None):
    lib().lux_gpu_push_chunk_scatter(
        _u64(stream), ctypes.c_int(is_min), ctypes.c_int(new_dense),
        dp(items), dp(counter), _u32(max_items), dp(push_row_ptr),
        dp(push_col), dp(old_labels), dp(snapshot), dp(new_labels),
        _u32(my_row_left), dp(new_seg), _u32(capacity), dp(visited_bits))


def publish_labels_guarded(stream, vp, meta, labels_part, labels_slice):
    lib().lux_gpu_publish_labels_guarded(_u64(stream), _u32(vp), dp(meta),
                                         dp(labels_part), dp(labels_slice))


def bits_from_labels(stream, vp, labels, bits):
    lib().lux_gpu_bits_from_labels(_u64(stream), _u32(vp), dp(labels),
                                   dp(bits))


def uf_union_star(stream, nv, star, parent):
    lib().lux_gpu_uf_union_star(_u64(stream), _u32(nv), dp(star), dp(parent))


def uf_flatten(stream, nv, parent, labels):
    lib().lux_gpu_uf_flatten(_u64(stream), _u32(nv), dp(parent), dp(labels))


def uf_union_binned(stream, n0, bin0, n1, bin1, n2, bin2, row_ptr, col,
                    row_left, parent, gbits=None):
    lib().lux_gpu_uf_union_binned(_u64(stream), _u32(n0), dp(bin0),
                                  _u32(n1), dp(bin1), _u32(n2), dp(bin2),
                                  dp(row_ptr), dp(col), _u32(row_left),
                                  dp(parent), dp(gbits))


def uf_union_kth(stream, vp, row_ptr, col, row_left, parent, k):
    lib().lux_gpu_uf_union_kth(_u64(stream), _u32(vp), dp(row_ptr), dp(col),
                               _u32(row_left), dp(parent), _u32(k))


def cc_giant_bits(stream, nv, labels, giant, bits):
    lib().lux_gpu_cc_giant_bits(_u64(stream), _u32(nv), dp(labels),
                                _u32(giant), dp(bits))
