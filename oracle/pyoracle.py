"""ctypes wrapper over liboracle.so — TEST INFRASTRUCTURE ONLY.

Per DESIGN.md §2 only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this module. The product path must never route
through it.
"""

import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))

c_u64 = ctypes.c_uint64
c_i64 = ctypes.c_int64
c_u32 = ctypes.c_uint32
c_i32 = ctypes.c_int32
c_vp = ctypes.c_void_p

_lib = None
_ref = None


def load():
    global _lib
    if _lib is None:
        p = os.path.join(_DIR, "liboracle.so")
        if not os.path.exists(p):
            raise RuntimeError(f"{p} missing — run `make -C oracle`")
        _lib = ctypes.CDLL(p)
        _decl(_lib)
    return _lib


def load_ref():
    """The reference-header shim (oracle/_ref/ref.so), or None if not built."""
    global _ref
    if _ref is None:
        p = os.path.join(_DIR, "_ref", "ref.so")
        if not os.path.exists(p):
            return None
        _ref = ctypes.CDLL(p)
        for f in ("ref_crc_hash_32", "ref_fnv_hash", "ref_xorshift32"):
            getattr(_ref, f).restype = c_u32
    return _ref


def _decl(lib):
    u = c_u32
    lib.orc_crc_hash_32.restype = u
    lib.orc_crc_hash_32.argtypes = [c_vp, c_i32, u]
    lib.orc_fnv_hash.restype = u
    lib.orc_fnv_hash.argtypes = [c_vp, c_i32, u]
    lib.orc_xorshift32.restype = u
    lib.orc_xorshift32.argtypes = [u]
    lib.orc_join_hash_u32.restype = u
    lib.orc_join_hash_u32.argtypes = [u, u]
    lib.orc_join_hash_u64.restype = u
    lib.orc_join_hash_u64.argtypes = [c_u64, u]
    lib.orc_join_hash_slice.restype = u
    lib.orc_join_hash_slice.argtypes = [c_vp, c_i32, u]
    lib.orc_calc_bucket_size.restype = u
    lib.orc_calc_bucket_size.argtypes = [u]
    lib.orc_join_select_key_constructor.restype = c_i32
    lib.orc_join_select_key_constructor.argtypes = [c_i32, c_vp, c_vp, c_i32, c_vp]
    lib.orc_join_select_varchar_constructor.restype = c_i32
    lib.orc_join_select_varchar_constructor.argtypes = [c_i32, c_i32]
    lib.orc_join_select_method.restype = c_i32
    lib.orc_join_select_method.argtypes = [c_i32, c_i32, c_u64, c_i64, c_i64, c_i32,
                                           c_i32, c_i32, c_i32, c_u64, c_u64]
    lib.orc_set_threads.argtypes = [c_i32]
    lib.orc_rle_page_encode_i32.restype = c_u64
    lib.orc_rle_page_encode_i32.argtypes = [c_vp, u, c_vp]
    lib.orc_rle_page_decode_i32.restype = c_u64
    lib.orc_rle_page_decode_i32.argtypes = [c_vp, c_vp]
    lib.orc_rle_page_encode_bool.restype = c_u64
    lib.orc_rle_page_encode_bool.argtypes = [c_vp, u, c_vp]
    lib.orc_rle_page_decode_bool.restype = c_u64
    lib.orc_rle_page_decode_bool.argtypes = [c_vp, c_vp]
    lib.orc_for_page_encode_i32.restype = c_u64
    lib.orc_for_page_encode_i32.argtypes = [c_vp, u, c_vp]
    lib.orc_for_page_decode_i32.restype = c_u64
    lib.orc_for_page_decode_i32.argtypes = [c_vp, c_u64, c_vp]
    lib.orc_binary_plain_encode.restype = c_u64
    lib.orc_binary_plain_encode.argtypes = [c_vp, c_vp, u, c_vp]
    lib.orc_binary_plain_decode.restype = c_u64
    lib.orc_binary_plain_decode.argtypes = [c_vp, c_u64, c_vp, c_vp]
    lib.orc_binary_prefix_encode.restype = c_u64
    lib.orc_binary_prefix_encode.argtypes = [c_vp, c_vp, u, c_vp]
    lib.orc_binary_prefix_decode.restype = c_u64
    lib.orc_binary_prefix_decode.argtypes = [c_vp, c_u64, c_vp, c_vp]
    lib.orc_plain_page_encode_i32.restype = c_u64
    lib.orc_plain_page_encode_i32.argtypes = [c_vp, u, c_vp]
    lib.orc_plain_page_decode_i32.restype = c_u64
    lib.orc_plain_page_decode_i32.argtypes = [c_vp, c_u64, c_vp]
    lib.orc_asof_inner_join.restype = None
    lib.orc_asof_inner_join.argtypes = [c_vp, c_vp, u, c_vp, c_vp, c_u64, c_i32, c_vp]
    lib.orc_asof_inner_join_nulls.restype = None
    lib.orc_asof_inner_join_nulls.argtypes = [c_vp, c_vp, c_vp, u, c_vp, c_vp, c_vp,
                                              c_u64, c_i32, c_vp]
    lib.orc_xxh3_64_4to8.restype = c_u64
    lib.orc_xxh3_64_4to8.argtypes = [c_vp, c_i32, c_u64]
    lib.orc_xxh3_hash_i32.argtypes = [c_vp, c_u64, c_vp]
    lib.orc_xxh3_hash_i64.argtypes = [c_vp, c_u64, c_vp]
    lib.orc_partition_channel_xxh3_u32.argtypes = [c_vp, c_u64, u, c_vp]
    lib.orc_zlib_crc32.restype = u
    lib.orc_zlib_crc32.argtypes = [c_vp, c_i32, u]
    lib.orc_partition_channel_crc_u32.argtypes = [c_vp, c_u64, u, c_vp]
    lib.orc_partition_channel_fnv_slice.argtypes = [c_vp, c_vp, c_u64, u, c_vp]
    lib.orc_gen_u64.restype = c_u64
    lib.orc_gen_u64.argtypes = [c_u64, c_u64, c_u64]
    lib.orc_gen_dates.argtypes = [c_i32, c_vp, c_vp]
    lib.orc_bucket_chained_build_u32.argtypes = [c_vp, u, c_vp, c_vp, u, u]
    lib.orc_bucket_chained_lookup_u32.argtypes = [c_vp, u, c_vp, u, u, c_vp]
    lib.orc_linear_chained_build_u32.argtypes = [c_vp, u, c_vp, c_vp, u, u]
    lib.orc_linear_chained_lookup_u32.argtypes = [c_vp, c_vp, u, c_vp, u, u, c_vp]
    lib.orc_range_direct_build_i32.argtypes = [c_vp, u, c_i64, c_vp, c_vp]
    lib.orc_range_direct_lookup_i32.argtypes = [c_vp, c_u64, c_i64, c_i64, c_vp, c_vp]
    lib.orc_bucket_chained_build_u64.argtypes = [c_vp, u, c_vp, c_vp, u, u]
    lib.orc_bucket_chained_lookup_u64.argtypes = [c_vp, u, c_vp, u, u, c_vp]
    lib.orc_probe_emit_u64.restype = c_u64
    lib.orc_probe_emit_u64.argtypes = [c_vp, c_vp, c_vp, c_vp, u, c_i32, c_vp, c_vp]
    lib.orc_slice_build_u32.argtypes = [c_vp, c_vp, u, c_vp, c_vp, u, u]
    lib.orc_slice_probe_emit.restype = c_u64
    lib.orc_slice_probe_emit.argtypes = [c_vp, c_vp, c_vp, u, c_vp, c_vp, c_vp, u, c_vp, c_vp]
    lib.orc_phmap_mix8.restype = c_u64
    lib.orc_phmap_mix8.argtypes = [c_u64]
    lib.orc_sbf_log_num_buckets.restype = c_i32
    lib.orc_sbf_log_num_buckets.argtypes = [c_u64]
    lib.orc_sbf_build_i32.argtypes = [c_vp, c_u64, c_vp, c_i32]
    lib.orc_sbf_test_i32.argtypes = [c_vp, c_u64, c_vp, c_i32, c_vp]
    lib.orc_eval_conjuncts_i64.restype = c_u64
    lib.orc_eval_conjuncts_i64.argtypes = [c_vp, c_i32, c_u64, c_vp, c_vp, c_vp, c_vp,
                                           c_i32]
    lib.orc_eval_conjuncts_i32.restype = c_u64
    lib.orc_eval_conjuncts_i32.argtypes = [c_vp, c_i32, c_u64, c_vp, c_vp, c_vp, c_vp,
                                           c_i32]
    lib.orc_dict_decode_binary.restype = c_u64
    lib.orc_dict_decode_binary.argtypes = [c_vp, c_vp, c_vp, c_u64, c_vp, c_vp]
    lib.orc_slice_build_nulls_u32.argtypes = [c_vp, c_vp, c_vp, u, c_vp, c_vp, u, u]
    lib.orc_slice_probe_emit_nulls.restype = c_u64
    lib.orc_slice_probe_emit_nulls.argtypes = [c_vp, c_vp, c_vp, u, c_vp, c_vp, c_vp, c_vp,
                                               u, c_i32, c_vp, c_vp]
    lib.orc_slice_probe_right.restype = c_u64
    lib.orc_slice_probe_right.argtypes = [c_vp, c_vp, c_vp, u, c_vp, u, c_vp, c_vp, u,
                                          c_i32, c_vp]
    lib.orc_slice_probe_emit_mode.restype = c_u64
    lib.orc_slice_probe_emit_mode.argtypes = [c_vp, c_vp, c_vp, u, c_vp, c_vp, c_vp, u,
                                              c_i32, c_vp, c_vp]
    lib.orc_probe_emit_u32.restype = c_u64
    lib.orc_probe_emit_u32.argtypes = [c_vp, c_vp, c_vp, c_vp, u, c_i32, c_vp, c_vp]
    lib.orc_filter_i64_lt.restype = c_u64
    lib.orc_filter_i64_lt.argtypes = [c_vp, c_u64, c_i64, c_vp]
    lib.orc_filter_i64_lt_mt.restype = c_u64
    lib.orc_filter_i64_lt_mt.argtypes = [c_vp, c_u64, c_i64, c_vp]
    lib.orc_partition_channel_u32.argtypes = [c_vp, c_u64, u, c_vp]
    lib.orc_partition_channel_2xi32.argtypes = [c_vp, c_vp, c_u64, u, c_vp]
    lib.orc_partition_channel_u64.argtypes = [c_vp, c_u64, u, c_vp]
    lib.orc_partition_counting_sort.argtypes = [c_vp, c_u64, u, c_vp, c_vp]
    lib.orc_q1_pipeline.restype = c_i64
    lib.orc_q1_pipeline.argtypes = [c_u64, c_u64, c_u64, c_i32, c_i32, ctypes.POINTER(c_u64)]
    lib.orc_q21_pipeline.argtypes = [c_u64, c_u64, c_u64, c_i32, c_i32, c_i32, c_vp]
    lib.orc_gen_lineorder_q1.argtypes = [c_u64, c_u64, c_u64, c_vp, c_vp, c_vp]
    lib.orc_gen_lineorder_q21.argtypes = [c_u64, c_u64, c_u64, c_vp, c_vp, c_vp, c_vp]
    lib.orc_gen_lineorder_q43.argtypes = [c_u64, c_u64, c_u64] + [c_vp] * 6
    lib.orc_build_cust_dim_q43.argtypes = [c_u64, u, c_i32, c_vp]
    lib.orc_build_supp_dim_q43.argtypes = [c_u64, u, c_i32, c_vp]
    lib.orc_build_part_dim_q43.argtypes = [c_u64, u, c_i32, c_vp]
    lib.orc_q43_kernel.argtypes = [c_vp] * 6 + [c_u64] + [c_vp] * 4 + [c_i64, c_i32, c_vp]
    lib.orc_q43_pipeline.argtypes = [c_u64, c_u64, c_u64, c_i32, c_i32, c_i32, c_i32, c_vp]
    lib.orc_gen_lineitem_q3.argtypes = [c_u64, c_u64, c_u64, c_u64] + [c_vp] * 4
    lib.orc_gen_orders_q3.argtypes = [c_u64, c_u64, u, c_vp, c_vp]
    lib.orc_gen_cust_mkt16.argtypes = [c_u64, u, c_vp]
    lib.orc_mkt_segment_literal.restype = ctypes.c_char_p
    lib.orc_mkt_segment_literal.argtypes = [c_i32]
    lib.orc_q3_build_cust_bits.argtypes = [c_vp, u, ctypes.c_char_p, c_vp]
    lib.orc_q3_build_order_bits.argtypes = [c_vp, c_vp, c_u64, c_vp, c_i32, c_vp]
    lib.orc_q3_probe_agg.restype = c_u64
    lib.orc_q3_probe_agg.argtypes = [c_vp] * 4 + [c_u64, c_vp, c_i32, c_vp, c_vp, c_u64]
    lib.orc_hash_agg_stats_u64.restype = c_u64
    lib.orc_hash_agg_stats_u64.argtypes = [c_vp, c_vp, c_u64] + [c_vp] * 5 + [c_u64]
    lib.orc_hash_agg_sum128_u64.restype = c_u64
    lib.orc_hash_agg_sum128_u64.argtypes = [c_vp, c_vp, c_u64] + [c_vp] * 3 + [c_u64]
    lib.orc_hash_agg_sum_u64.restype = c_u64
    lib.orc_hash_agg_sum_u64.argtypes = [c_vp, c_vp, c_u64, c_vp, c_vp, c_vp, c_u64]
    lib.orc_q1_kernel.restype = c_i64
    lib.orc_q1_kernel.argtypes = [c_vp, c_vp, c_vp, c_u64, c_vp, c_i64, c_i64, c_i32,
                                  ctypes.POINTER(c_u64)]
    lib.orc_q21_kernel.argtypes = [c_vp, c_vp, c_vp, c_vp, c_u64, c_vp, c_vp, c_vp,
                                   c_i64, c_i32, c_vp]


def _p(a: np.ndarray):
    return a.ctypes.data_as(c_vp)


def crc_hash_32(data: bytes, seed: int) -> int:
    return load().orc_crc_hash_32(data, len(data), seed)


def fnv_hash(data: bytes, seed: int) -> int:
    return load().orc_fnv_hash(data, len(data), seed)


def join_hash_slice(data: bytes, num_buckets: int) -> int:
    return load().orc_join_hash_slice(data, len(data), num_buckets)


def slice_join(bbytes, boffsets, row_count, pbytes, poffsets, probe_rows, max_out):
    """Build a Slice-key chained table over build rows 1..row_count and emit
    all (probe_idx, build_idx) pairs. bbytes/pbytes uint8 arrays;
    boffsets uint32[row_count+2] (row 0 = empty sentinel), poffsets
    uint32[probe_rows+1]."""
    lib = load()
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size).bit_length() - 1
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_slice_build_u32(_p(bbytes), _p(boffsets), row_count, _p(first), _p(nxt),
                            bucket_size, log)
    op = np.empty(max_out, np.uint32)
    ob = np.empty(max_out, np.uint32)
    m = lib.orc_slice_probe_emit(_p(bbytes), _p(boffsets), _p(nxt), bucket_size, _p(first),
                                 _p(pbytes), _p(poffsets), probe_rows, _p(op), _p(ob))
    return op[:m], ob[:m]


def slice_join_mode(bbytes, boffsets, row_count, pbytes, poffsets, probe_rows, mode,
                    max_out):
    """Per-join-type Slice probe: mode 0 INNER, 1 SEMI, 2 ANTI, 3 OUTER."""
    lib = load()
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size).bit_length() - 1
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_slice_build_u32(_p(bbytes), _p(boffsets), row_count, _p(first), _p(nxt),
                            bucket_size, log)
    op = np.empty(max_out, np.uint32)
    ob = np.empty(max_out, np.uint32)
    m = lib.orc_slice_probe_emit_mode(_p(bbytes), _p(boffsets), _p(nxt), bucket_size,
                                      _p(first), _p(pbytes), _p(poffsets), probe_rows,
                                      mode, _p(op), _p(ob))
    return op[:m], ob[:m]


def q1_pipeline(seed, row_start, n_rows, year, threads=0):
    cnt = c_u64()
    s = load().orc_q1_pipeline(seed, row_start, n_rows, year, threads, ctypes.byref(cnt))
    return s, cnt.value


def q21_pipeline(seed, row_start, n_rows, category, region, threads=0) -> np.ndarray:
    out = np.zeros(7000, dtype=np.int64)
    load().orc_q21_pipeline(seed, row_start, n_rows, category, region, threads, _p(out))
    return out


def filter_i64_lt(inp: np.ndarray, theta: int, mt=False) -> np.ndarray:
    out = np.empty_like(inp)
    fn = load().orc_filter_i64_lt_mt if mt else load().orc_filter_i64_lt
    k = fn(_p(inp), len(inp), theta, _p(out))
    return out[:k]


def gen_lineorder_q1(seed, row_start, n):
    od = np.empty(n, np.int32)
    ep = np.empty(n, np.int32)
    dc = np.empty(n, np.int32)
    load().orc_gen_lineorder_q1(seed, row_start, n, _p(od), _p(ep), _p(dc))
    return od, ep, dc


def gen_lineorder_q21(seed, row_start, n):
    pk = np.empty(n, np.int32)
    sk = np.empty(n, np.int32)
    od = np.empty(n, np.int32)
    rv = np.empty(n, np.int32)
    load().orc_gen_lineorder_q21(seed, row_start, n, _p(pk), _p(sk), _p(od), _p(rv))
    return pk, sk, od, rv


def q1_kernel(od, ep, dc, dfirst, mn, mx, threads=0):
    cnt = c_u64()
    s = load().orc_q1_kernel(_p(od), _p(ep), _p(dc), len(od), _p(dfirst), mn, mx,
                             threads, ctypes.byref(cnt))
    return s, cnt.value


def q21_kernel(pk, sk, od, rv, pfirst, sfirst, dfirst, dmin, threads=0):
    out = np.zeros(7000, np.int64)
    load().orc_q21_kernel(_p(pk), _p(sk), _p(od), _p(rv), len(pk), _p(pfirst),
                          _p(sfirst), _p(dfirst), dmin, threads, _p(out))
    return out


def gen_lineorder_q43(seed, row_start, n):
    cols = [np.empty(n, np.int32) for _ in range(6)]
    load().orc_gen_lineorder_q43(seed, row_start, n, *[_p(c) for c in cols])
    return cols  # ck, sk, pk, od, rv, sc


def q43_pipeline(seed, row_start, n_rows, region=1, nation=7, category=12, threads=0):
    out = np.zeros(800, np.int64)
    load().orc_q43_pipeline(seed, row_start, n_rows, region, nation, category,
                            threads, _p(out))
    return out


def q43_kernel(ck, sk, pk, od, rv, sc, cfirst, sfirst, pfirst, dfirst, dmin, threads=0):
    out = np.zeros(800, np.int64)
    load().orc_q43_kernel(_p(ck), _p(sk), _p(pk), _p(od), _p(rv), _p(sc), len(ck),
                          _p(cfirst), _p(sfirst), _p(pfirst), _p(dfirst), dmin,
                          threads, _p(out))
    return out


def gen_lineitem_q3(seed, row_start, n, n_orders):
    lk = np.empty(n, np.int64)
    ext = np.empty(n, np.int64)
    disc = np.empty(n, np.int64)
    ship = np.empty(n, np.int32)
    load().orc_gen_lineitem_q3(seed, row_start, n, n_orders, _p(lk), _p(ext), _p(disc), _p(ship))
    return lk, ext, disc, ship


def gen_orders_q3(seed, n_orders, n_custs):
    oc = np.empty(n_orders, np.int32)
    od = np.empty(n_orders, np.int32)
    load().orc_gen_orders_q3(seed, n_orders, n_custs, _p(oc), _p(od))
    return oc, od


def gen_cust_mkt16(seed, n_custs):
    out = np.empty(n_custs * 16, np.uint8)
    load().orc_gen_cust_mkt16(seed, n_custs, _p(out))
    return out


def mkt_literal(idx=1) -> bytes:
    return load().orc_mkt_segment_literal(idx)


def q3_pipeline(seed, row_start, n, n_orders, n_custs, seg=1,
                date_cutoff=19950315, ship_cutoff=19950315):
    """Full Q3 oracle pipeline; returns (keys, sums) key-sorted."""
    mkt = gen_cust_mkt16(seed, n_custs)
    cbits = np.zeros((n_custs + 7) // 8, np.uint8)
    load().orc_q3_build_cust_bits(_p(mkt), n_custs, mkt_literal(seg), _p(cbits))
    oc, od = gen_orders_q3(seed, n_orders, n_custs)
    obits = np.zeros((n_orders + 7) // 8, np.uint8)
    load().orc_q3_build_order_bits(_p(oc), _p(od), n_orders, _p(cbits), date_cutoff, _p(obits))
    lk, ext, disc, ship = gen_lineitem_q3(seed, row_start, n, n_orders)
    cap = max(n, 16)
    ok = np.empty(cap, np.uint64)
    os_ = np.empty(cap, np.int64)
    g = load().orc_q3_probe_agg(_p(lk), _p(ext), _p(disc), _p(ship), n, _p(obits),
                                ship_cutoff, _p(ok), _p(os_), cap)
    assert g != 2**64 - 1
    order = np.argsort(ok[:g])
    return ok[:g][order].copy(), os_[:g][order].copy()


def bshuf_lz4_encode_i32(values: np.ndarray):
    lib = load()
    lib.orc_bshuf_lz4_encode_i32.restype = c_u64
    lib.orc_bshuf_lz4_encode_i32.argtypes = [c_vp, c_u32, c_vp, c_vp]
    out = np.zeros(len(values) * 4 + 16 * (len(values) // 2048 + 2), np.uint8)
    n = lib.orc_bshuf_lz4_encode_i32(_p(values), len(values), _p(out), None)
    return out[:n].copy()


def bshuf_lz4_decode_i32(page: np.ndarray, n: int):
    lib = load()
    lib.orc_bshuf_lz4_decode_i32.restype = c_u64
    lib.orc_bshuf_lz4_decode_i32.argtypes = [c_vp, c_u32, c_vp]
    vals = np.zeros(n, np.int32)
    used = lib.orc_bshuf_lz4_decode_i32(_p(page), n, _p(vals))
    assert used == len(page), (used, len(page))
    return vals


def hash_agg_stats(keys, vals):
    cap = max(len(keys), 16)
    o = [np.empty(cap, np.uint64)] + [np.empty(cap, np.int64) for _ in range(4)]
    g = load().orc_hash_agg_stats_u64(_p(keys), _p(vals), len(keys), *[_p(x) for x in o], cap)
    assert g != 2**64 - 1
    return tuple(x[:g].copy() for x in o)


def hash_agg_sum128(keys, vals):
    cap = max(len(keys), 16)
    ok = np.empty(cap, np.uint64)
    lo = np.empty(cap, np.uint64)
    hi = np.empty(cap, np.int64)
    g = load().orc_hash_agg_sum128_u64(_p(keys), _p(vals), len(keys), _p(ok), _p(lo), _p(hi), cap)
    assert g != 2**64 - 1
    return ok[:g].copy(), lo[:g].copy(), hi[:g].copy()


def hash_agg_sum(keys: np.ndarray, vals: np.ndarray):
    cap = max(len(keys), 16)
    ok = np.empty(cap, np.uint64)
    os_ = np.empty(cap, np.int64)
    oc = np.empty(cap, np.int64)
    g = load().orc_hash_agg_sum_u64(_p(keys), _p(vals), len(keys), _p(ok), _p(os_), _p(oc), cap)
    assert g != 2**64 - 1
    return ok[:g].copy(), os_[:g].copy(), oc[:g].copy()


def range_direct_build(keys_1based: np.ndarray, min_value: int, max_value: int):
    """keys_1based[0] is the sentinel row. Returns (first, next)."""
    row_count = len(keys_1based) - 1
    interval = max_value - min_value + 1
    first = np.zeros(interval, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    load().orc_range_direct_build_i32(_p(keys_1based), row_count, min_value, _p(first), _p(nxt))
    return first, nxt


def range_direct_lookup(probe_keys: np.ndarray, min_value, max_value, first: np.ndarray):
    heads = np.zeros(len(probe_keys), np.uint32)
    load().orc_range_direct_lookup_i32(_p(probe_keys), len(probe_keys), min_value, max_value,
                                       _p(first), _p(heads))
    return heads


def bucket_chained_build_nulls(keys_1based, nulls_1based):
    lib = load()
    lib.orc_bucket_chained_build_nulls_u32.argtypes = [c_vp, c_vp, c_u32, c_vp, c_vp,
                                                       c_u32, c_u32]
    row_count = len(keys_1based) - 1
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size - 1).bit_length()
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_bucket_chained_build_nulls_u32(_p(keys_1based), _p(nulls_1based), row_count,
                                           _p(first), _p(nxt), bucket_size, log)
    return first, nxt, bucket_size, log


def bucket_chained_lookup_nulls(probe_keys, probe_nulls, first, bucket_size, log):
    lib = load()
    lib.orc_bucket_chained_lookup_nulls_u32.argtypes = [c_vp, c_vp, c_u32, c_vp, c_u32,
                                                        c_u32, c_vp]
    heads = np.zeros(len(probe_keys), np.uint32)
    lib.orc_bucket_chained_lookup_nulls_u32(_p(probe_keys), _p(probe_nulls),
                                            len(probe_keys), _p(first), bucket_size, log,
                                            _p(heads))
    return heads


def pack_keys_2xi32(a, b):
    lib = load()
    lib.orc_pack_keys_2xi32.argtypes = [c_vp, c_vp, c_u64, c_vp]
    out = np.zeros(len(a), np.uint64)
    lib.orc_pack_keys_2xi32(_p(a), _p(b), len(a), _p(out))
    return out


def probe_emit_mode(build_keys, nxt, probe_keys, heads, mode):
    lib = load()
    lib.orc_probe_emit_mode_u32.restype = c_u64
    lib.orc_probe_emit_mode_u32.argtypes = [c_vp, c_vp, c_vp, c_vp, c_u32, c_i32, c_vp, c_vp]
    cap = max(len(probe_keys) * 8, 1024)
    op = np.empty(cap, np.uint32)
    ob = np.empty(cap, np.uint32)
    m = lib.orc_probe_emit_mode_u32(_p(build_keys), _p(nxt), _p(probe_keys), _p(heads),
                                    len(probe_keys), mode, _p(op), _p(ob))
    return op[:m].copy(), ob[:m].copy()


def probe_right(build_keys, nxt, probe_keys, heads, anti):
    lib = load()
    lib.orc_probe_right_u32.restype = c_u64
    lib.orc_probe_right_u32.argtypes = [c_vp, c_vp, c_u32, c_vp, c_vp, c_u32, c_i32, c_vp]
    out = np.empty(len(build_keys), np.uint32)
    m = lib.orc_probe_right_u32(_p(build_keys), _p(nxt), len(build_keys) - 1,
                                _p(probe_keys), _p(heads), len(probe_keys), anti, _p(out))
    return out[:m].copy()


def probe_emit(build_keys: np.ndarray, nxt: np.ndarray, probe_keys: np.ndarray,
               heads: np.ndarray, collision_free=False):
    cap = max(len(probe_keys) * 8, 1024)
    op = np.empty(cap, np.uint32)
    ob = np.empty(cap, np.uint32)
    m = load().orc_probe_emit_u32(_p(build_keys), _p(nxt), _p(probe_keys), _p(heads),
                                  len(probe_keys), 1 if collision_free else 0, _p(op), _p(ob))
    return op[:m].copy(), ob[:m].copy()


def bucket_chained_build(keys_1based: np.ndarray):
    row_count = len(keys_1based) - 1
    bucket_size = load().orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size - 1).bit_length()
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    load().orc_bucket_chained_build_u32(_p(keys_1based), row_count, _p(first), _p(nxt),
                                        bucket_size, log)
    return first, nxt, bucket_size, log


def bucket_chained_lookup(probe_keys: np.ndarray, first: np.ndarray, bucket_size, log):
    heads = np.zeros(len(probe_keys), np.uint32)
    load().orc_bucket_chained_lookup_u32(_p(probe_keys), len(probe_keys), _p(first),
                                         bucket_size, log, _p(heads))
    return heads


def linear_chained_build(keys_1based: np.ndarray):
    row_count = len(keys_1based) - 1
    bucket_size = load().orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size - 1).bit_length()
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    load().orc_linear_chained_build_u32(_p(keys_1based), row_count, _p(first), _p(nxt),
                                        bucket_size, log)
    return first, nxt, bucket_size, log


def linear_chained_lookup(build_keys: np.ndarray, probe_keys: np.ndarray, first, bucket_size, log):
    heads = np.zeros(len(probe_keys), np.uint32)
    load().orc_linear_chained_lookup_u32(_p(build_keys), _p(probe_keys), len(probe_keys),
                                         _p(first), bucket_size, log, _p(heads))
    return heads


def partition_channels(keys: np.ndarray, num_channels: int) -> np.ndarray:
    out = np.zeros(len(keys), np.uint32)
    load().orc_partition_channel_u32(_p(keys), len(keys), num_channels, _p(out))
    return out


def partition_channels_u64(keys: np.ndarray, num_channels: int) -> np.ndarray:
    out = np.zeros(len(keys), np.uint32)
    load().orc_partition_channel_u64(_p(keys), len(keys), num_channels, _p(out))
    return out


def partition_counting_sort(channel_ids: np.ndarray, num_channels: int):
    sp = np.zeros(num_channels + 1, np.uint64)
    ri = np.zeros(len(channel_ids), np.uint32)
    load().orc_partition_counting_sort(_p(channel_ids), len(channel_ids), num_channels,
                                       _p(sp), _p(ri))
    return sp, ri


def sbf_build(keys_i32: np.ndarray):
    """SimdBlockFilter build (runtime_filter.cpp:26-36): returns (directory,
    log_num_buckets)."""
    lib = load()
    log = lib.orc_sbf_log_num_buckets(len(keys_i32))
    directory = np.zeros((1 << log) * 8, np.uint32)
    lib.orc_sbf_build_i32(_p(keys_i32), len(keys_i32), _p(directory), log)
    return directory, log


def sbf_test(keys_i32: np.ndarray, directory: np.ndarray, log: int) -> np.ndarray:
    out = np.zeros(len(keys_i32), np.uint8)
    load().orc_sbf_test_i32(_p(keys_i32), len(keys_i32), _p(directory), log, _p(out))
    return out


def slice_join_nulls(bbytes, boffsets, bnulls, row_count, pbytes, poffsets, pnulls,
                     probe_rows, mode, max_out):
    """Nullable Slice-key join: null build rows never chain; null probe rows
    match nothing (ANTI/OUTER emit them unmatched)."""
    lib = load()
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size).bit_length() - 1
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_slice_build_nulls_u32(_p(bbytes), _p(boffsets), _p(bnulls), row_count,
                                  _p(first), _p(nxt), bucket_size, log)
    op = np.empty(max_out, np.uint32)
    ob = np.empty(max_out, np.uint32)
    m = lib.orc_slice_probe_emit_nulls(_p(bbytes), _p(boffsets), _p(nxt), bucket_size,
                                       _p(first), _p(pbytes), _p(poffsets), _p(pnulls),
                                       probe_rows, mode, _p(op), _p(ob))
    return op[:m], ob[:m]


def dict_decode_binary(dict_bytes, dict_offsets, codes):
    """binary_dict_page.cpp:229-280 — codes -> BinaryColumn (bytes, offsets)."""
    n = len(codes)
    lens = dict_offsets[np.asarray(codes) + 1] - dict_offsets[np.asarray(codes)]
    out_bytes = np.zeros(max(int(lens.sum()), 1), np.uint8)
    out_offsets = np.zeros(n + 1, np.uint32)
    total = load().orc_dict_decode_binary(_p(dict_bytes), _p(dict_offsets),
                                          _p(np.ascontiguousarray(codes, np.int32)), n,
                                          _p(out_bytes), _p(out_offsets))
    return out_bytes[:total], out_offsets


def bucket_chained_join_u64(build_keys_1based, probe_keys, max_out):
    """8-byte-key chained join (JoinKeyHash<8>): emit all match pairs."""
    lib = load()
    row_count = len(build_keys_1based) - 1
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size - 1).bit_length()
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_bucket_chained_build_u64(_p(build_keys_1based), row_count, _p(first), _p(nxt),
                                     bucket_size, log)
    heads = np.zeros(len(probe_keys), np.uint32)
    lib.orc_bucket_chained_lookup_u64(_p(probe_keys), len(probe_keys), _p(first),
                                      bucket_size, log, _p(heads))
    op = np.empty(max_out, np.uint32)
    ob = np.empty(max_out, np.uint32)
    m = lib.orc_probe_emit_u64(_p(build_keys_1based), _p(nxt), _p(probe_keys), _p(heads),
                               len(probe_keys), 0, _p(op), _p(ob))
    return op[:m], ob[:m]


def slice_probe_right(bbytes, boffsets, row_count, pbytes, poffsets, probe_rows, anti):
    """RIGHT SEMI/ANTI over Slice keys: matched/unmatched build rows."""
    lib = load()
    bucket_size = lib.orc_calc_bucket_size(row_count + 1)
    log = int(bucket_size).bit_length() - 1
    first = np.zeros(bucket_size, np.uint32)
    nxt = np.zeros(row_count + 1, np.uint32)
    lib.orc_slice_build_u32(_p(bbytes), _p(boffsets), row_count, _p(first), _p(nxt),
                            bucket_size, log)
    out = np.empty(row_count, np.uint32)
    m = lib.orc_slice_probe_right(_p(bbytes), _p(boffsets), _p(nxt), bucket_size, _p(first),
                                  row_count, _p(pbytes), _p(poffsets), probe_rows, anti,
                                  _p(out))
    return np.sort(out[:m])


def eval_conjuncts(cols, preds):
    """Eager-prune multi-conjunct filter (chunk_predicate_evaluator.cpp:31-80).
    cols: list of int32 arrays (modified in place); preds: (col,op,lo,hi)."""
    import ctypes as ct
    n = len(cols[0])
    ptrs = (c_vp * len(cols))(*[c.ctypes.data_as(c_vp).value for c in cols])
    pc = np.array([p[0] for p in preds], np.int32)
    po = np.array([p[1] for p in preds], np.int32)
    pl = np.array([p[2] for p in preds], np.int32)
    ph = np.array([p[3] for p in preds], np.int32)
    m = load().orc_eval_conjuncts_i32(ptrs, len(cols), n, _p(pc), _p(po), _p(pl), _p(ph),
                                      len(preds))
    return m


def eval_conjuncts_i64(cols, preds):
    n = len(cols[0])
    ptrs = (c_vp * len(cols))(*[c.ctypes.data_as(c_vp).value for c in cols])
    pc = np.array([p[0] for p in preds], np.int32)
    po = np.array([p[1] for p in preds], np.int32)
    pl = np.array([p[2] for p in preds], np.int64)
    ph = np.array([p[3] for p in preds], np.int64)
    return load().orc_eval_conjuncts_i64(ptrs, len(cols), n, _p(pc), _p(po), _p(pl),
                                         _p(ph), len(preds))


# --- JoinHashMapSelector restatement (join_hash_table.cpp:164-344) ---
JM_NAMES = {0: "DIRECT", 1: "RANGE_DIRECT", 2: "RANGE_DIRECT_SET",
            3: "DENSE_RANGE_DIRECT", 4: "LINEAR_CHAINED",
            5: "LINEAR_CHAINED_SET", 6: "BUCKET_CHAINED"}
KEYCON_NAMES = {0: "ONE_KEY", 1: "ONE_KEY_VARCHAR", 2: "FIXED_INT",
                3: "FIXED_BIGINT", 4: "FIXED_LARGEINT", 5: "SERIALIZED_VARCHAR"}


def join_select_key_constructor(fixed_sizes, null_safe=None,
                                enable_fixed_size_string=1):
    n = len(fixed_sizes)
    fs = np.asarray(fixed_sizes, np.int32)
    ns = np.asarray(null_safe if null_safe is not None else [0] * n, np.uint8)
    pb = c_i32()
    kc = load().orc_join_select_key_constructor(n, _p(fs), _p(ns),
                                                enable_fixed_size_string,
                                                ctypes.byref(pb))
    return kc, pb.value


def join_select_varchar_constructor(max_size, enable_fixed_size_string=1):
    return load().orc_join_select_varchar_constructor(max_size, enable_fixed_size_string)


def join_select_method(key_constructor, lt_class, row_count, min_value, max_value,
                       mode=0, with_other_conjunct=0, enable_range_direct=1,
                       enable_linear_chained=1, l2_size=0, l3_size=0):
    return load().orc_join_select_method(key_constructor, lt_class, row_count,
                                         min_value, max_value, mode,
                                         with_other_conjunct, enable_range_direct,
                                         enable_linear_chained, l2_size, l3_size)


XXH3_SEED_32 = 0x9E3779B1


def xxh3_64(data: bytes, seed: int = 0) -> int:
    """XXH3-64 for 4-8 byte inputs (the exchange hash's per-value call)."""
    return load().orc_xxh3_64_4to8(data, len(data), seed)


def xxh3_exchange_hash_i32(cols) -> np.ndarray:
    """Version-1 exchange hash over i32 key columns: seed XXH3_SEED_32,
    chained per column, u32 truncation per hop
    (exchange_sink_operator.cpp:604-610)."""
    n = len(cols[0])
    h = np.full(n, XXH3_SEED_32, np.uint32)
    for col in cols:
        c = np.ascontiguousarray(col, np.int32)
        load().orc_xxh3_hash_i32(_p(c), n, _p(h))
    return h


def partition_channels_xxh3(keys: np.ndarray, num_channels: int) -> np.ndarray:
    ch = np.empty(len(keys), np.uint32)
    load().orc_partition_channel_xxh3_u32(_p(np.ascontiguousarray(keys, np.uint32)),
                                          len(keys), num_channels, _p(ch))
    return ch


def rle_page_encode_i32(values: np.ndarray) -> np.ndarray:
    """RLE page (rle_page.h header + rle_encoding.h hybrid at bit_width 32)."""
    a = np.ascontiguousarray(values, np.int32)
    out = np.zeros(4 + a.nbytes + len(a) // 8 + 64, np.uint8)
    nb = load().orc_rle_page_encode_i32(_p(a), len(a), _p(out))
    return out[:nb].copy()


def rle_page_decode_i32(page: np.ndarray, n: int) -> np.ndarray:
    v = np.zeros(n, np.int32)
    load().orc_rle_page_decode_i32(_p(np.ascontiguousarray(page, np.uint8)), _p(v))
    return v


def rle_page_encode_bool(values: np.ndarray) -> np.ndarray:
    a = np.ascontiguousarray(values, np.uint8)
    out = np.zeros(4 + len(a) + len(a) // 8 + 64, np.uint8)
    nb = load().orc_rle_page_encode_bool(_p(a), len(a), _p(out))
    return out[:nb].copy()


def rle_page_decode_bool(page: np.ndarray, n: int) -> np.ndarray:
    v = np.zeros(n, np.uint8)
    load().orc_rle_page_decode_bool(_p(np.ascontiguousarray(page, np.uint8)), _p(v))
    return v


def for_page_encode_i32(values: np.ndarray) -> np.ndarray:
    a = np.ascontiguousarray(values, np.int32)
    out = np.zeros(a.nbytes * 2 + 64, np.uint8)
    nb = load().orc_for_page_encode_i32(_p(a), len(a), _p(out))
    return out[:nb].copy()


def for_page_decode_i32(page: np.ndarray, n: int) -> np.ndarray:
    v = np.zeros(n, np.int32)
    got = load().orc_for_page_decode_i32(_p(np.ascontiguousarray(page, np.uint8)),
                                         len(page), _p(v))
    assert got == n
    return v


def binary_plain_encode(bytes_: np.ndarray, offsets: np.ndarray) -> np.ndarray:
    n = len(offsets) - 1
    out = np.zeros(int(offsets[-1]) + 4 * n + 8, np.uint8)
    nb = load().orc_binary_plain_encode(_p(np.ascontiguousarray(bytes_, np.uint8)),
                                        _p(np.ascontiguousarray(offsets, np.uint32)),
                                        n, _p(out))
    return out[:nb].copy()


def binary_plain_decode(page: np.ndarray, n: int):
    pg = np.ascontiguousarray(page, np.uint8)
    body = len(pg) - 4 - 4 * n
    b = np.zeros(max(body, 1), np.uint8)
    o = np.zeros(n + 1, np.uint32)
    got = load().orc_binary_plain_decode(_p(pg), len(pg), _p(b), _p(o))
    assert got == n
    return b[:body], o


def zlib_crc32(data: bytes, seed: int = 0) -> int:
    return load().orc_zlib_crc32(data, len(data), seed)


def partition_channels_crc(keys: np.ndarray, num_channels: int) -> np.ndarray:
    ch = np.empty(len(keys), np.uint32)
    load().orc_partition_channel_crc_u32(_p(np.ascontiguousarray(keys, np.uint32)),
                                         len(keys), num_channels, _p(ch))
    return ch


def partition_channels_fnv_slice(bytes_: np.ndarray, offsets: np.ndarray,
                                 num_channels: int) -> np.ndarray:
    n = len(offsets) - 1
    ch = np.empty(n, np.uint32)
    load().orc_partition_channel_fnv_slice(
        _p(np.ascontiguousarray(bytes_, np.uint8)),
        _p(np.ascontiguousarray(offsets, np.uint32)), n, num_channels, _p(ch))
    return ch


def binary_prefix_encode(bytes_: np.ndarray, offsets: np.ndarray) -> np.ndarray:
    n = len(offsets) - 1
    out = np.zeros(int(offsets[-1]) * 2 + n * 12 + 64, np.uint8)
    nb = load().orc_binary_prefix_encode(
        _p(np.ascontiguousarray(bytes_, np.uint8)),
        _p(np.ascontiguousarray(offsets, np.uint32)), n, _p(out))
    return out[:nb].copy()


def binary_prefix_decode(page: np.ndarray, n: int, total_bytes: int):
    pg = np.ascontiguousarray(page, np.uint8)
    b = np.zeros(max(total_bytes, 1), np.uint8)
    o = np.zeros(n + 1, np.uint32)
    got = load().orc_binary_prefix_decode(_p(pg), len(pg), _p(b), _p(o))
    assert got == n
    return b[:int(o[-1])], o


def asof_inner_join(build_keys, build_asof, probe_keys, probe_asof,
                    opcode: int) -> np.ndarray:
    """AsofIndex restatement (join_hash_table_descriptor.cpp:70-134).
    build_* are 1-based (index 0 = sentinel); returns per-probe-row matched
    1-based build row (0 = miss). opcode: 0 LT / 1 LE / 2 GT / 3 GE."""
    bk = np.ascontiguousarray(build_keys, np.int32)
    ba = np.ascontiguousarray(build_asof, np.int64)
    pk = np.ascontiguousarray(probe_keys, np.int32)
    pa = np.ascontiguousarray(probe_asof, np.int64)
    assert len(bk) == len(ba) and len(pk) == len(pa)
    out = np.empty(len(pk), np.uint32)
    load().orc_asof_inner_join(_p(bk), _p(ba), len(bk) - 1, _p(pk), _p(pa),
                               len(pk), opcode, _p(out))
    return out


def asof_inner_join_nulls(build_keys, build_asof, build_nulls, probe_keys,
                          probe_asof, probe_nulls, opcode: int) -> np.ndarray:
    """Nulls variant (join_hash_table_descriptor.h:447-456): flagged build
    rows skipped, null probe rows unmatched. Either mask may be None."""
    bk = np.ascontiguousarray(build_keys, np.int32)
    ba = np.ascontiguousarray(build_asof, np.int64)
    pk = np.ascontiguousarray(probe_keys, np.int32)
    pa = np.ascontiguousarray(probe_asof, np.int64)
    bn = None if build_nulls is None else np.ascontiguousarray(build_nulls, np.uint8)
    pn = None if probe_nulls is None else np.ascontiguousarray(probe_nulls, np.uint8)
    out = np.empty(len(pk), np.uint32)
    load().orc_asof_inner_join_nulls(
        _p(bk), _p(ba), None if bn is None else _p(bn), len(bk) - 1,
        _p(pk), _p(pa), None if pn is None else _p(pn), len(pk), opcode, _p(out))
    return out


def plain_page_encode_i32(values) -> np.ndarray:
    v = np.ascontiguousarray(values, np.int32)
    out = np.zeros(4 + v.nbytes, np.uint8)
    nb = load().orc_plain_page_encode_i32(_p(v), len(v), _p(out))
    return out[:nb].copy()


def plain_page_decode_i32(page, n: int) -> np.ndarray:
    pg = np.ascontiguousarray(page, np.uint8)
    out = np.zeros(max(n, 1), np.int32)
    got = load().orc_plain_page_decode_i32(_p(pg), len(pg), _p(out))
    assert got == n, got
    return out[:n]
